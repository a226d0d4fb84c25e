set -x
cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/r2_pytest.log 2>&1
echo "pytest rc=$?"
timeout 240 python bench.py --steps 30 --warmup 5 > gpurun_out/r2_bench1.json 2>&1
echo "bench rc=$?"
timeout 300 python main.py --train-flag --synthetic --synthetic-size 512 --batch-size 16 --amp --end-epoch 2 --print-interval 8 --num-workers 8 --save-path /tmp/w_graph > gpurun_out/r2_cli_graph.log 2>&1
echo "cli_graph rc=$?"
timeout 300 python main.py --train-flag --synthetic --synthetic-size 512 --batch-size 16 --amp --end-epoch 2 --print-interval 8 --num-workers 8 --no-train-graph --save-path /tmp/w_eager > gpurun_out/r2_cli_eager.log 2>&1
echo "cli_eager rc=$?"
timeout 240 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29511 bench.py --gpus 2 --steps 10 --warmup 3 --batch-size 8 > gpurun_out/r2_rccl2.log 2>&1
echo "rccl2 rc=$?"
tail -3 gpurun_out/r2_pytest.log
tail -2 gpurun_out/r2_bench1.json
tail -4 gpurun_out/r2_cli_graph.log
tail -4 gpurun_out/r2_cli_eager.log
tail -6 gpurun_out/r2_rccl2.log
