#!/bin/bash
# Round-2 GPU call #4: GBT device-loop A/B, flagship host-side profile,
# boosting GPU tier re-check.
set -x
mkdir -p gpurun_out
export PYTHONPATH="$PWD"

timeout 300 python tools/boost_prof.py 2>&1 | tee gpurun_out/r4_boost_time.log
timeout 600 python -m pytest tests/test_boosting_gpu.py tests/test_forest_gpu.py -x -q 2>&1 | tee gpurun_out/r4_boost_tier.log
timeout 900 python tools/host_profile.py 2>&1 | tee gpurun_out/r4_host_profile.log
