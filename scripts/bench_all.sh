#!/bin/bash
# All BASELINE.json shapes on one GPU (see bench.py for multi-GPU launch)
set -e
cd "$(dirname "$0")/.."
for cfg in nell2 netflix delicious4d amazon; do
  python bench.py --config "$cfg" --steps 10 --warmup 3
done
