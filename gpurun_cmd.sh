cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 700 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu.log 2>&1
echo "PG_RC=$?"; tail -1 gpurun_out/pytest_gpu.log
