export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== build check (extension fresh) =="
python -c "from skdist_amd.ops.build import extension_is_stale; print('stale:', extension_is_stale())"
echo "== full gpu suite =="
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -2
echo "== graft smoke =="
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1
echo "== bench =="
timeout 700 python bench.py --steps 3 --warmup 1 2>/dev/null | tail -1 | tee gpurun_out/bench_final.json | python3 -c "import json,sys; d=json.load(sys.stdin); print(round(d['ms_per_step'],1),'ms/step', round(d['value']),'fits/s', 'vs_baseline', round(d['vs_baseline'],1))"
echo "== non-gpu tests on gpu box (should all pass/skip cleanly) =="
timeout 900 python -m pytest tests -m "not gpu" -q -x 2>&1 | tail -1
