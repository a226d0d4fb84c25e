cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/r2s_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/r2s_pytest.log
timeout 240 python bench.py --steps 30 --warmup 5 > gpurun_out/r2s_bench1.json 2>&1
echo train:; tail -1 gpurun_out/r2s_bench1.json
timeout 240 python bench.py --steps 20 --warmup 5 --num-stack 2 --increase-ch 128 > gpurun_out/r2s_big.json 2>&1
echo big:; tail -1 gpurun_out/r2s_big.json
