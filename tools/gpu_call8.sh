#!/bin/bash
# Round-2 GPU call #8: sparse tier incl. renorm test; textscale after
# sorted-gen + sync-free epochs.
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"

timeout 600 python -m pytest tests/test_sparse_gpu.py -x -q 2>&1 | tee gpurun_out/r8_sparse_tier.log
timeout 900 python tools/textscale_bench.py --n 1000000 --holdout 50000 \
    --candidates 40 --folds 5 --epochs 10 --batch-size 1024 \
    2>&1 | tee gpurun_out/r8_textscale.log
timeout 900 python tools/textscale_bench.py --n 1000000 --holdout 50000 \
    --candidates 40 --folds 5 --epochs 10 --batch-size 1024 \
    2>&1 | tee -a gpurun_out/r8_textscale.log
