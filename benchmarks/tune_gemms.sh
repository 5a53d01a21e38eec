#!/bin/bash
# Regenerate the hipBLASLt TunableOp cache for gfx950 on an MI355X box.
# Run from the repo root; results land in config/tunableop/gfx950_*.csv.
set -e
export PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1
export PYTORCH_TUNABLEOP_FILENAME=config/tunableop/gfx950_.csv
mkdir -p config/tunableop
python bench.py --steps 4 --warmup 2
python bench.py --phase 2 --steps 4 --warmup 2
# replicate the device-0 cache for all 8 local devices
for d in 1 2 3 4 5 6 7; do cp config/tunableop/gfx950_0.csv config/tunableop/gfx950_${d}.csv; done
