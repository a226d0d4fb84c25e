set -x
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 1200 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "PYTEST_RC=$?"; tail -2 gpurun_out/pytest_gpu.log
timeout 600 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_train.json 2> gpurun_out/bench_train.log
echo "BT_RC=$?"; cat gpurun_out/bench_train.json
HSA_ENABLE_IPC_MODE_LEGACY=0 timeout 600 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 --master-addr 127.0.0.1 --master-port 29517 bench.py --gpus 1 --steps 5 --warmup 2 > gpurun_out/bench_torchrun.log 2>&1
echo "TR_RC=$?"; tail -1 gpurun_out/bench_torchrun.log
timeout 600 python bench.py --mode infer --batch-size 1 --steps 100 --warmup 20 --graph > gpurun_out/bench_infer_b1.json 2> gpurun_out/bi1.log
echo "BI1_RC=$?"; cat gpurun_out/bench_infer_b1.json
