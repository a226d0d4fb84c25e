cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/r2d_pytest.log 2>&1
echo "pytest rc=$?"; tail -2 gpurun_out/r2d_pytest.log
timeout 120 python tools/tr_probe.py > gpurun_out/r2d_trprobe.log 2>&1
echo "trprobe rc=$?"
timeout 120 python tools/kbench.py stem --iters 40 > gpurun_out/r2d_kbench_stem.log 2>&1
cat gpurun_out/r2d_kbench_stem.log | grep -v libdrm
timeout 240 python bench.py --steps 30 --warmup 5 > gpurun_out/r2d_bench1.json 2>&1
tail -1 gpurun_out/r2d_bench1.json
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof_r2d -o prof -- python /root/repo/bench.py --steps 5 --warmup 2 --no-train-graph > /root/repo/gpurun_out/r2d_prof.log 2>&1
echo "prof rc=$?"; ls /root/repo/gpurun_out/prof_r2d/ 2>/dev/null
