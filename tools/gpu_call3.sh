#!/bin/bash
# Round-2 GPU call #3: full GPU tier with all round-2 additions, measured
# covtype-scale numbers, colsum-fix A/B re-profile, GBT profile, flagship
# bench sanity, and SQ counters on the flagship SGD kernels (headroom gate).
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"
PROF=/tmp/prof_out   # big rocprof dbs stay OUT of gpurun_out (64 MiB cap)
mkdir -p "$PROF"

timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tee gpurun_out/r3_gpu_tier.log

timeout 600 python examples/search/covtype_scale.py 2>&1 | tee gpurun_out/r3_covtype.log
timeout 600 python examples/eliminate/covtype.py 2>&1 | tee gpurun_out/r3_covtype_elim.log

timeout 600 python bench.py --gpus 1 --steps 5 --warmup 2 2>&1 | tee gpurun_out/r3_bench.log

cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$PROF/sparse2" -- \
    python "$GRAFT_REPO_ROOT/tools/textscale_bench.py" --n 200000 --holdout 20000 \
    --candidates 8 --folds 5 --epochs 10 \
    > "$GRAFT_REPO_ROOT/gpurun_out/r3_sparse_prof.log" 2>&1

timeout 600 rocprofv3 --kernel-trace --stats -d "$PROF/boost" -- \
    python "$GRAFT_REPO_ROOT/tools/boost_prof.py" \
    > "$GRAFT_REPO_ROOT/gpurun_out/r3_boost_prof.log" 2>&1

# PMC pass (own run, counters only — no trace domains)
timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
    SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES \
    -d "$PROF/sgd_pmc" -- \
    python "$GRAFT_REPO_ROOT/bench.py" --gpus 1 --steps 2 --warmup 1 \
    > "$GRAFT_REPO_ROOT/gpurun_out/r3_sgd_pmc.log" 2>&1
# summarize on the box; only the text summaries travel back
cd "$GRAFT_REPO_ROOT"
for db in "$PROF"/sparse2/*/*.db; do
  python tools/prof_summary.py "$db" > gpurun_out/r3_sparse2_summary.txt 2>&1
done
for db in "$PROF"/boost/*/*.db; do
  python tools/prof_summary.py "$db" > gpurun_out/r3_boost_summary.txt 2>&1
done
ls "$PROF"/sgd_pmc/ > gpurun_out/r3_pmc_ls.txt 2>&1
for db in "$PROF"/sgd_pmc/*/*.db; do
  python tools/prof_summary.py "$db" > gpurun_out/r3_sgd_pmc_summary.txt 2>&1
done
du -sh gpurun_out >> gpurun_out/r3_pmc_ls.txt
echo "ALL DONE rc=$?"
