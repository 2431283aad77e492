#!/bin/bash
# Round-2 GPU call #3: full GPU tier with all round-2 additions, measured
# covtype-scale numbers, colsum-fix A/B re-profile, GBT profile, flagship
# bench sanity, and SQ counters on the flagship SGD kernels (headroom gate).
set -x
mkdir -p gpurun_out

timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tee gpurun_out/r3_gpu_tier.log

timeout 600 python examples/search/covtype_scale.py 2>&1 | tee gpurun_out/r3_covtype.log
timeout 600 python examples/eliminate/covtype.py 2>&1 | tee gpurun_out/r3_covtype_elim.log

timeout 600 python bench.py --gpus 1 --steps 5 --warmup 2 2>&1 | tee gpurun_out/r3_bench.log

cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof_sparse2" -- \
    python "$GRAFT_REPO_ROOT/tools/textscale_bench.py" --n 200000 --holdout 20000 \
    --candidates 8 --folds 5 --epochs 10 \
    > "$GRAFT_REPO_ROOT/gpurun_out/r3_sparse_prof.log" 2>&1

timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof_boost" -- \
    python "$GRAFT_REPO_ROOT/tools/boost_prof.py" \
    > "$GRAFT_REPO_ROOT/gpurun_out/r3_boost_prof.log" 2>&1

# PMC pass (own run, counters only — no trace domains)
timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
    SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES \
    -d "$GRAFT_REPO_ROOT/gpurun_out/prof_sgd_pmc" -- \
    python "$GRAFT_REPO_ROOT/bench.py" --gpus 1 --steps 2 --warmup 1 \
    > "$GRAFT_REPO_ROOT/gpurun_out/r3_sgd_pmc.log" 2>&1
echo "ALL DONE rc=$?"
