cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/r2g_pytest.log 2>&1
echo "pytest rc=$?"; tail -2 gpurun_out/r2g_pytest.log
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph --fp8 > gpurun_out/r2g_infer_fp8.json 2>&1
echo fp8:; tail -1 gpurun_out/r2g_infer_fp8.json
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph > gpurun_out/r2g_infer_b8.json 2>&1
echo bf16:; tail -1 gpurun_out/r2g_infer_b8.json
timeout 600 python tools/quality_probe.py holdout --steps 3000 --train-imgs 256 --val-imgs 64 --eval-every 500 > gpurun_out/r2g_holdout.log 2>&1
echo "holdout rc=$?"; grep -v libdrm gpurun_out/r2g_holdout.log | head -8
