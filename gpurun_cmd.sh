cd /root/repo
timeout 1450 python main.py --train-flag --synthetic --synthetic-size 7581 --batch-size 16 --amp --end-epoch 100 --num-workers 12 --print-interval 400 --save-path /tmp/full > gpurun_out/final_fulltrain.log 2>&1
echo "train rc=$?"; tail -2 gpurun_out/final_fulltrain.log
timeout 300 python main.py --synthetic --synthetic-size 1000 --random-seed 4242 --model-load /tmp/full/check_point_100.pth --save-path /tmp/full --conf-th 0.15 > gpurun_out/final_fulleval.log 2>&1
echo "eval rc=$?"; grep -iE 'map|ap ' gpurun_out/final_fulleval.log | tail -5
