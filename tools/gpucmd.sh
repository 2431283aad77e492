export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== build freshness ==" && python -c "from skdist_amd.ops.build import extension_is_stale as s; print('stale:', s())"
echo "== gpu suite ==" && timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -1
echo "== cpu suite on gpu box ==" && timeout 1200 python -m pytest tests -m "not gpu" -q 2>&1 | tail -1
echo "== smoke ==" && timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1
echo "== bench ==" && timeout 700 python bench.py --steps 3 --warmup 1 2>/dev/null | tail -1 | python3 -c "import json,sys; d=json.load(sys.stdin); print(round(d['ms_per_step'],1),'ms/step', round(d['value']),'fits/s')"
echo "== examples on GPU (device paths engage) =="
for f in examples/search/basic_usage.py examples/ensemble/basic_usage.py examples/eliminate/basic_usage.py examples/predict/basic_usage.py; do
  PYTHONPATH=/root/repo timeout 300 python $f > /dev/null 2>&1 && echo "OK $f" || echo "FAIL $f"
done
