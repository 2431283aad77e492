#!/bin/bash
# Round-2 GPU call #10: grid-stride sparse kernel A/B + tier re-check.
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"
timeout 600 python -m pytest tests/test_sparse_gpu.py -x -q 2>&1 | tee gpurun_out/r10_sparse_tier.log
timeout 900 python tools/textscale_bench.py --n 1000000 --holdout 50000 \
    --candidates 40 --folds 5 --epochs 10 --batch-size 1024 \
    2>&1 | tee gpurun_out/r10_textscale.log
timeout 900 python tools/textscale_bench.py --n 1000000 --holdout 50000 \
    --candidates 40 --folds 5 --epochs 10 --batch-size 1024 \
    2>&1 | tee -a gpurun_out/r10_textscale.log
PROF=/tmp/prof_out; mkdir -p "$PROF"
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$PROF/sparse3" -- \
    python "$GRAFT_REPO_ROOT/tools/textscale_bench.py" --n 200000 --holdout 20000 \
    --candidates 8 --folds 5 --epochs 10 \
    > "$GRAFT_REPO_ROOT/gpurun_out/r10_sparse_prof.log" 2>&1
cd "$GRAFT_REPO_ROOT"
for db in "$PROF"/sparse3/*/*.db; do
  python tools/prof_summary.py "$db" > gpurun_out/r10_sparse3_summary.txt 2>&1
done
