set -x
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "PG_RC=$?"; tail -2 gpurun_out/pytest_gpu.log
timeout 600 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_train.json 2> gpurun_out/bench_train.log
echo "BT_RC=$?"; cat gpurun_out/bench_train.json
