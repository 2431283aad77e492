#!/bin/bash
# Round-2 GPU call #7: full tier after the broadcast-skip change, bench,
# textscale host-side cProfile.
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tee gpurun_out/r7_gpu_tier.log
timeout 600 python bench.py --gpus 1 --steps 10 --warmup 3 2>&1 | tee gpurun_out/r7_bench.log

timeout 900 python - <<'PYEOF' 2>&1 | tee gpurun_out/r7_textscale_prof.log
import cProfile, pstats, sys, os
sys.path.insert(0, os.getcwd())
sys.argv = ["x", "--n", "1000000", "--holdout", "50000", "--candidates",
            "40", "--folds", "5", "--epochs", "10", "--batch-size", "1024"]
from tools.textscale_bench import main
main()  # warm
pr = cProfile.Profile()
pr.enable(); main(); pr.disable()
pstats.Stats(pr).sort_stats("cumulative").print_stats(30)
PYEOF
