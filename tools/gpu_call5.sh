#!/bin/bash
# Round-2 GPU call #5: verify the host-path wins on the flagship bench,
# re-run the FULL gpu tier, re-measure textscale with the K2b fix.
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"

timeout 600 python bench.py --gpus 1 --steps 10 --warmup 3 2>&1 | tee gpurun_out/r5_bench.log
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tee gpurun_out/r5_gpu_tier.log
timeout 900 python tools/textscale_bench.py --n 1000000 --holdout 50000 \
    --candidates 40 --folds 5 --epochs 10 --batch-size 1024 \
    2>&1 | tee gpurun_out/r5_textscale.log
