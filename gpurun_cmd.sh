cd /root/repo
echo "=== full reference recipe: 100 epochs, 7581 synthetic imgs, batch 16, AMP, milestones [50,90] ==="
timeout 1450 python main.py --train-flag --synthetic --synthetic-size 7581 --batch-size 16 --amp --end-epoch 100 --num-workers 12 --print-interval 200 --save-path /tmp/full > gpurun_out/final_fulltrain.log 2>&1
echo "train rc=$?"; tail -3 gpurun_out/final_fulltrain.log
ls /tmp/full/check_point_100.pth 2>/dev/null || ls /tmp/full | tail -2
timeout 300 python main.py --synthetic --synthetic-size 1000 --random-seed 424242 --model-load /tmp/full/check_point_100.pth --save-path /tmp/full --conf-th 0.15 > gpurun_out/final_fulleval.log 2>&1
echo "eval rc=$?"; grep -i 'map' gpurun_out/final_fulleval.log | tail -3
