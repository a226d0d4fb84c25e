cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 700 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "PG_RC=$?"; tail -1 gpurun_out/pytest_gpu.log
timeout 240 python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE-OK')" 2>&1 | tail -1
timeout 400 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_train.json 2> gpurun_out/bench_train.log
echo "BT_RC=$?"; cat gpurun_out/bench_train.json
