#!/bin/bash
# GPU numerics suite — run on an MI355X (parity: reference scripts/mpi_test.sh)
set -e
cd "$(dirname "$0")/.."
python -m pytest tests -q -m gpu "$@"
