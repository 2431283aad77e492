export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== numerics with BN=256 kernels =="
timeout 600 python -m pytest tests/test_gpu_kernels.py -x -q > gpurun_out/pytest_k.log 2>&1; echo "rc=$?"
tail -2 gpurun_out/pytest_k.log
run_bench () {
  cp skdist_amd/ops/variants/$1 skdist_amd/ops/_skdist_hip.so
  timeout 700 python bench.py --steps 3 --warmup 1 2>/dev/null | tail -1 | python3 -c "import json,sys; d=json.load(sys.stdin); print('$1', round(d['ms_per_step'],1), 'ms/step', round(d['value']), 'fits/s')"
}
echo "== same-box A/B 128 vs 256 tiles =="
run_bench hip_bn128.so
run_bench hip_bn256.so
run_bench hip_bn128.so
run_bench hip_bn256.so
cp skdist_amd/ops/variants/hip_bn256.so skdist_amd/ops/_skdist_hip.so
echo "== stats for BN=256 =="
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof5 -o prof5 -- python /root/repo/bench.py --steps 1 --warmup 1 --epochs 6 > /root/repo/gpurun_out/prof5.log 2>&1
cd /root/repo
python tools/prof_summary.py gpurun_out/prof5/prof5_results.db 2>&1 | head -6
