set -x
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 600 python -m pytest tests/test_gpu_kernels.py -x -q > gpurun_out/pytest_k.log 2>&1
echo "P_RC=$?"; tail -1 gpurun_out/pytest_k.log
timeout 300 python tools/kbench.py bn --iters 30 > gpurun_out/kb_bn.log 2>&1
grep -v amdgpu gpurun_out/kb_bn.log
timeout 600 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_train.json 2> gpurun_out/bench_train.log
echo "BT_RC=$?"; cat gpurun_out/bench_train.json
