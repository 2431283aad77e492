#!/bin/bash
# Round-2 GPU call #11: hipGraph epoch replay A/B (bench + textscale) +
# full tier.
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tee gpurun_out/r11_gpu_tier.log
timeout 600 python bench.py --gpus 1 --steps 10 --warmup 3 2>&1 | tee gpurun_out/r11_bench.log
timeout 600 python bench.py --gpus 1 --steps 10 --warmup 3 2>&1 | tee -a gpurun_out/r11_bench.log
timeout 900 python tools/textscale_bench.py --n 1000000 --holdout 50000 \
    --candidates 40 --folds 5 --epochs 10 --batch-size 1024 \
    2>&1 | tee gpurun_out/r11_textscale.log
