cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 600 python -m pytest tests/test_gpu_kernels.py tests/test_gpu_e2e.py -x -q -k "stem or wgrad or backward or determin" > gpurun_out/pytest_k.log 2>&1
echo "P_RC=$?"; tail -1 gpurun_out/pytest_k.log
timeout 300 python tools/kbench.py stem_wgrad --iters 20 > gpurun_out/kb_s.log 2>&1
grep -v amdgpu gpurun_out/kb_s.log | tail -3
timeout 600 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_train.json 2> gpurun_out/bench_train.log
echo "BT_RC=$?"; cat gpurun_out/bench_train.json
