#!/bin/bash
# CPU test suite (parity: reference scripts/test.sh)
set -e
cd "$(dirname "$0")/.."
python -m pytest tests -q -m "not gpu" "$@"
