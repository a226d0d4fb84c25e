cd /root/repo
timeout 240 python bench.py --steps 30 --warmup 5 > gpurun_out/r2t_bench1.json 2>&1
echo train:; tail -1 gpurun_out/r2t_bench1.json
timeout 240 python bench.py --steps 30 --warmup 5 --no-train-graph > gpurun_out/r2t_nograph.json 2>&1
echo nograph:; tail -1 gpurun_out/r2t_nograph.json
python -m pytest tests/test_gpu_kernels.py -x -q > gpurun_out/r2t_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/r2t_pytest.log
