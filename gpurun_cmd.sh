cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 300 python main.py --train-flag --synthetic --synthetic-size 16 --amp \
  --batch-size 8 --end-epoch 1 --num-workers 2 --imsize 256 \
  --num-stack 1 --hourglass-inch 64 --print-interval 1 \
  --save-path gpurun_out/cli_run > gpurun_out/cli_train.log 2>&1
echo "TRAIN_RC=$?"; tail -2 gpurun_out/cli_train.log
ls gpurun_out/cli_run/ | head -5
timeout 200 python main.py --synthetic --synthetic-size 4 --num-workers 0 \
  --imsize 256 --save-path gpurun_out/cli_run \
  --model-load gpurun_out/cli_run/check_point_1.pth > gpurun_out/cli_eval.log 2>&1
echo "EVAL_RC=$?"; tail -2 gpurun_out/cli_eval.log
