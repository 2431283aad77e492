export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== gpu kernel numerics =="
timeout 600 python -m pytest tests/test_gpu_kernels.py -x -q > gpurun_out/pytest_k.log 2>&1; echo "rc=$?"
tail -2 gpurun_out/pytest_k.log
echo "== bench =="
timeout 700 python bench.py --steps 3 --warmup 1 2>/dev/null | tail -1 | tee gpurun_out/bench_opt3.json
echo "== kernel stats =="
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof4 -o prof4 -- python /root/repo/bench.py --steps 1 --warmup 1 --epochs 6 > /root/repo/gpurun_out/prof4.log 2>&1; echo "rc=$?"
cd /root/repo
python tools/prof_summary.py gpurun_out/prof4/prof4_results.db 2>&1 | head -6
