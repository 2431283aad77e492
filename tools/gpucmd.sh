export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== gpu suite ==" && timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -1
echo "== smoke ==" && timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1
echo "== bench (final) ==" && timeout 700 python bench.py --steps 5 --warmup 1 2>/dev/null | tail -1 | tee gpurun_out/bench_roundend.json | python3 -c "import json,sys; d=json.load(sys.stdin); print(round(d['ms_per_step'],1),'ms/step', round(d['value']),'fits/s', 'vs_baseline', round(d['vs_baseline'],1), 'best', round(d['config']['best_score'],5))"
