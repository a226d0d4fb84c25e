set -x
cd /tmp
export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
# final kernel-stats profile (current state)
timeout 900 rocprofv3 --kernel-trace --stats -d /tmp/prof2 -- python /root/repo/bench.py --steps 5 --warmup 2 --no-train-graph > /root/repo/gpurun_out/prof2.log 2>&1
echo "PROF_RC=$?"
cp -r /tmp/prof2 /root/repo/gpurun_out/prof2 2>/dev/null
cd /root/repo
timeout 300 python tools/kbench.py stem_wgrad --iters 20 > gpurun_out/kb_s2.log 2>&1
grep -v amdgpu gpurun_out/kb_s2.log | tail -3
# fp8 inference evidence
timeout 600 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --fp8 > gpurun_out/bench_infer_fp8.json 2> gpurun_out/bif.log
echo "FP8_RC=$?"; cat gpurun_out/bench_infer_fp8.json
