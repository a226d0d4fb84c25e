cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 700 python -m pytest tests/test_gpu_kernels.py tests/test_gpu_e2e.py -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "PG_RC=$?"; tail -1 gpurun_out/pytest_gpu.log
timeout 300 python bench.py --steps 20 --warmup 8 2>/dev/null | tail -1
