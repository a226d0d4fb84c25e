cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/r2o_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/r2o_pytest.log
timeout 300 python bench.py --steps 20 --warmup 5 --num-stack 2 --increase-ch 128 > gpurun_out/r2o_big.json 2>&1
echo big:; tail -1 gpurun_out/r2o_big.json
timeout 300 python bench.py --steps 20 --warmup 5 --num-stack 2 > gpurun_out/r2o_2stack.json 2>&1
echo 2stack:; tail -1 gpurun_out/r2o_2stack.json
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof_b1 -o p -- python /root/repo/bench.py --mode infer --batch-size 1 --steps 50 --warmup 10 > /root/repo/gpurun_out/r2o_b1prof.log 2>&1
echo "b1 prof rc=$?"; tail -2 /root/repo/gpurun_out/r2o_b1prof.log | head -1
