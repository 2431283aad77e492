export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== full gpu suite =="
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -1
echo "== multimodel batched on GPU =="
timeout 600 python -m pytest tests/test_multimodel.py -q 2>&1 | tail -1
echo "== bench =="
timeout 700 python bench.py --steps 3 --warmup 1 2>/dev/null | tail -1 | python3 -c "import json,sys; d=json.load(sys.stdin); print(round(d['ms_per_step'],1),'ms/step', round(d['value']),'fits/s', 'best', round(d['config']['best_score'],4))"
echo "== smoke =="
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1
