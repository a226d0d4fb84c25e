cd /tmp
export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
timeout 500 python /root/repo/bench.py --steps 10 --warmup 3 --num-stack 2 --increase-ch 128 > /root/repo/gpurun_out/bench_big.json 2> /root/repo/gpurun_out/bench_big.log
echo "BIG_RC=$?"; cat /root/repo/gpurun_out/bench_big.json; tail -2 /root/repo/gpurun_out/bench_big.log
timeout 500 rocprofv3 --kernel-trace --stats -d /tmp/prof3 -- python /root/repo/bench.py --steps 4 --warmup 2 --no-train-graph > /root/repo/gpurun_out/prof3.log 2>&1
echo "PROF_RC=$?"
cp -r /tmp/prof3 /root/repo/gpurun_out/prof3 2>/dev/null
