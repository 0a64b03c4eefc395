from splatt_amd.cli import main
import sys
sys.exit(main())
