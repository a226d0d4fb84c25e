cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof_r2final -o prof -- python /root/repo/bench.py --steps 5 --warmup 2 --no-train-graph > /root/repo/gpurun_out/r2l_prof.log 2>&1
echo "prof rc=$?"
cd /root/repo
timeout 300 python bench.py --steps 20 --warmup 5 --num-stack 2 --increase-ch 128 > gpurun_out/r2l_bench_big.json 2>&1
echo big:; tail -1 gpurun_out/r2l_bench_big.json
timeout 120 python tools/kbench.py wgrad --iters 40 > gpurun_out/r2l_kbench_wgrad.log 2>&1
grep wgrad gpurun_out/r2l_kbench_wgrad.log
timeout 240 python bench.py --steps 30 --warmup 5 --dtype fp32 --no-train-graph > gpurun_out/r2l_bench_fp32.json 2>&1
echo fp32:; tail -1 gpurun_out/r2l_bench_fp32.json
