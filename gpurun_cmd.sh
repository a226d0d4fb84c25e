cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/r2m_pytest.log 2>&1
echo "pytest rc=$?"; tail -2 gpurun_out/r2m_pytest.log
timeout 240 python bench.py --steps 30 --warmup 5 > gpurun_out/r2m_bench1.json 2>&1
echo train:; tail -1 gpurun_out/r2m_bench1.json
timeout 300 python main.py --train-flag --synthetic --synthetic-size 512 --batch-size 16 --amp --end-epoch 2 --print-interval 8 --num-workers 8 --save-path /tmp/wg > gpurun_out/r2m_cli.log 2>&1
echo "cli rc=$?"; grep 'Loss' gpurun_out/r2m_cli.log | tail -3
