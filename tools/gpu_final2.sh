#!/bin/bash
# Closing dress rehearsal: driver-shaped tier + smoke + bench, plus the
# headline example measurements on the same box for the record.
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tee gpurun_out/rz_tier.log
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" 2>&1 | tee gpurun_out/rz_smoke.log
timeout 600 python bench.py --gpus 1 --steps 10 --warmup 3 2>&1 | tee gpurun_out/rz_bench.log
timeout 600 python examples/search/covtype_scale.py 2>&1 | tee gpurun_out/rz_covtype.log
timeout 900 python examples/postprocessing/text_voter.py 2>&1 | tee gpurun_out/rz_voter.log
timeout 600 python tools/textscale_bench.py --n 1000000 --holdout 50000 \
    --candidates 40 --folds 5 --epochs 10 --batch-size 1024 2>&1 | tee gpurun_out/rz_textscale.log
