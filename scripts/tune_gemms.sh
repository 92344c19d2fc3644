#!/bin/bash
# Run on an MI355X box: tunes hipBLASLt algorithm selection (TunableOp) for
# the flagship bench shapes; the table lands in gpurun_out/ for merge-back
# and is then committed as artifacts/tunableop_gfx950.csv.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
export PYTORCH_TUNABLEOP_ENABLED=1
export PYTORCH_TUNABLEOP_TUNING=1
export PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tunableop_gfx950.csv
export TOSEM_NOTUNE=1   # skip the read-only autoload branch
timeout 2400 python bench.py --steps 3 --warmup 2 "$@"
ls -la gpurun_out/tunableop*
