cd /root/repo
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph --fp8 > gpurun_out/r2i_infer_fp8.json 2>&1
echo fp8:; tail -1 gpurun_out/r2i_infer_fp8.json
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph > gpurun_out/r2i_infer_b8.json 2>&1
echo bf16:; tail -1 gpurun_out/r2i_infer_b8.json
timeout 180 python -m pytest tests/test_gpu_e2e.py -x -q > gpurun_out/r2i_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/r2i_pytest.log
timeout 180 python bench.py --mode infer --batch-size 1 --steps 100 --warmup 20 --graph > gpurun_out/r2i_infer_b1.json 2>&1
echo b1:; tail -1 gpurun_out/r2i_infer_b1.json
