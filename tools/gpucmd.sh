export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== pytest gpu =="
timeout 600 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1; echo "pytest rc=$?"
tail -5 gpurun_out/pytest_gpu.log
echo "== smoke =="
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" > gpurun_out/smoke.log 2>&1; echo "smoke rc=$?"
tail -3 gpurun_out/smoke.log
echo "== bench =="
timeout 700 python bench.py --steps 2 --warmup 1 > gpurun_out/bench1.log 2>&1; echo "bench rc=$?"
tail -3 gpurun_out/bench1.log
echo "== rocprof =="
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -o prof -- python /root/repo/bench.py --steps 1 --warmup 0 --candidates-per-gpu 100 > /root/repo/gpurun_out/prof.log 2>&1; echo "rocprof rc=$?"
tail -3 /root/repo/gpurun_out/prof.log
ls /root/repo/gpurun_out/prof 2>/dev/null | head
