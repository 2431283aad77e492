export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
cd /tmp
timeout 700 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof7 -o prof7 -- python /root/repo/bench.py --steps 3 --warmup 1 > /root/repo/gpurun_out/prof7.log 2>&1
echo "stats rc=$?"
timeout 700 rocprofv3 --kernel-trace --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES -d /root/repo/gpurun_out/pmc7 -o pmc7 -- python /root/repo/bench.py --steps 1 --warmup 1 > /root/repo/gpurun_out/pmc7.log 2>&1
echo "pmc rc=$?"
cd /root/repo
python tools/prof_summary.py gpurun_out/prof7/prof7_results.db 2>&1 | head -8
