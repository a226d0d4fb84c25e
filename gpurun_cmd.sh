cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 500 python -m pytest tests/test_gpu_kernels.py -x -q -k "wgrad or conv or stem" > gpurun_out/pytest_k.log 2>&1
echo "P_RC=$?"; tail -1 gpurun_out/pytest_k.log
timeout 300 python tools/kbench.py wgrad --iters 30 2>/dev/null | grep wgrad
timeout 300 python bench.py --steps 20 --warmup 8 2>/dev/null | tail -1
