set -x
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 900 python -m pytest tests/test_gpu_kernels.py tests/test_gpu_e2e.py -x -q -k "stem or wgrad or backward or overfit" > gpurun_out/pytest_s.log 2>&1
echo "P_RC=$?"; tail -2 gpurun_out/pytest_s.log
timeout 300 python tools/kbench.py stem_wgrad --iters 20 > gpurun_out/kb_s.log 2>&1
grep -v amdgpu gpurun_out/kb_s.log | tail -4
timeout 600 python bench.py --steps 20 --warmup 8 > gpurun_out/bench_train.json 2> gpurun_out/bench_train.log
echo "BT_RC=$?"; cat gpurun_out/bench_train.json
