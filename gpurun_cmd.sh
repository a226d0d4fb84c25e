cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/r2x_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/r2x_pytest.log
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph > gpurun_out/r2x_b8.json 2>&1
echo b8:; tail -1 gpurun_out/r2x_b8.json
timeout 180 python bench.py --mode infer --batch-size 1 --steps 100 --warmup 20 --graph > gpurun_out/r2x_b1.json 2>&1
echo b1:; tail -1 gpurun_out/r2x_b1.json
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph --fp8 > gpurun_out/r2x_fp8.json 2>&1
echo fp8:; tail -1 gpurun_out/r2x_fp8.json
