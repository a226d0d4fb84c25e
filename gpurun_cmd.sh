cd /root/repo
timeout 120 python tools/kbench.py wgrad --iters 40 > gpurun_out/r2n_wgrad.log 2>&1
grep wgrad gpurun_out/r2n_wgrad.log
timeout 240 python bench.py --steps 30 --warmup 5 > gpurun_out/r2n_bench1.json 2>&1
echo train:; tail -1 gpurun_out/r2n_bench1.json
python -m pytest tests/test_gpu_kernels.py tests/test_gpu_e2e.py -x -q > gpurun_out/r2n_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/r2n_pytest.log
