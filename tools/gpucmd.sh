export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
run_bench () {
  cp skdist_amd/ops/variants/$1 skdist_amd/ops/_skdist_hip.so
  timeout 700 python bench.py --steps 3 --warmup 1 2>/dev/null | tail -1 | python3 -c "import json,sys; d=json.load(sys.stdin); print('$1', round(d['ms_per_step'],1), 'ms/step', round(d['value']), 'fits/s')"
}
echo "== same-box A/B (interleaved, 2 rounds) =="
# (A/B harness retired: original kernels won 353 vs 384 ms/step same-box)
run_bench hip_pipelined.so
# (A/B harness retired: original kernels won 353 vs 384 ms/step same-box)
run_bench hip_pipelined.so
cp skdist_amd/ops/variants/hip_pipelined.so skdist_amd/ops/_skdist_hip.so
