export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /root/repo
echo "== quick regression: bench after scoring/refit opts =="
timeout 700 python bench.py --steps 2 --warmup 1 2>/dev/null | tail -1 | tee gpurun_out/bench_opt1.json
echo "== pmc counters on short bench =="
cd /tmp
timeout 900 rocprofv3 --kernel-trace --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES -d /root/repo/gpurun_out/pmc1 -o pmc1 -- python /root/repo/bench.py --steps 1 --warmup 1 --candidates-per-gpu 500 --epochs 4 > /root/repo/gpurun_out/pmc1.log 2>&1; echo "pmc rc=$?"
cd /root/repo
python tools/pmc_summary.py gpurun_out/pmc1 2>&1 | head -40 | tee gpurun_out/pmc1_summary.txt
