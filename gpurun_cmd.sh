cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 700 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "PG_RC=$?"; tail -1 gpurun_out/pytest_gpu.log
timeout 400 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_train.json 2>/dev/null
echo "BT_RC=$?"; cat gpurun_out/bench_train.json
timeout 300 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph > gpurun_out/bench_infer_b8.json 2>/dev/null
cat gpurun_out/bench_infer_b8.json
