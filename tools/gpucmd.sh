export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
<<<<<<< HEAD
echo "== gpu kernel numerics =="
timeout 600 python -m pytest tests/test_gpu_kernels.py tests/test_distributed_cpu.py -x -q > gpurun_out/pytest_k.log 2>&1; echo "rc=$?"
tail -2 gpurun_out/pytest_k.log
echo "== bench =="
timeout 700 python bench.py --steps 3 --warmup 1 2>/dev/null | tail -1 | tee gpurun_out/bench_opt2.json
echo "== kernel stats =="
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof3 -o prof3 -- python /root/repo/bench.py --steps 1 --warmup 1 --epochs 6 > /root/repo/gpurun_out/prof3.log 2>&1; echo "rc=$?"
cd /root/repo
python tools/prof_summary.py gpurun_out/prof3/prof3_results.db 2>&1 | head -10
=======
echo "== hash gpu tests =="
timeout 600 python -m pytest tests/test_hash_gpu.py -x -q > gpurun_out/pytest_hash.log 2>&1; echo "rc=$?"
tail -6 gpurun_out/pytest_hash.log
echo "== full gpu suite =="
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1; echo "rc=$?"
tail -2 gpurun_out/pytest_gpu.log
echo "== bench default =="
timeout 700 python bench.py --steps 2 --warmup 1 2>/dev/null | tail -1 | tee gpurun_out/bench_b8k.json
echo "== bench batch 16384 =="
timeout 700 python bench.py --steps 2 --warmup 1 --batch-size 16384 2>/dev/null | tail -1 | tee gpurun_out/bench_b16k.json
echo "== bench batch 32768 =="
timeout 700 python bench.py --steps 2 --warmup 1 --batch-size 32768 2>/dev/null | tail -1 | tee gpurun_out/bench_b32k.json
echo "== rocprof steady-state (steps 3 warmup 1) =="
cd /tmp
timeout 900 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof2 -o prof2 -- python /root/repo/bench.py --steps 3 --warmup 1 > /root/repo/gpurun_out/prof2.log 2>&1; echo "rocprof rc=$?"
cd /root/repo
python tools/prof_summary.py gpurun_out/prof2/prof2_results.db > gpurun_out/prof2_summary.txt 2>&1
head -25 gpurun_out/prof2_summary.txt
>>>>>>> parent of ac37f34 (SGD GEMMs: B operand via direct global 16B fragment loads instead of LDS staging (PMC showed glds-bandwidth bound))
