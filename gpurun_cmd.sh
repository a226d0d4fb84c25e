cd /root/repo
timeout 300 python main.py --train-flag --synthetic --synthetic-size 128 --batch-size 16 --amp --end-epoch 1 --print-interval 8 --num-workers 4 --save-path /tmp/w1 > /dev/null 2>&1
python3 - <<'PY'
from real_time_helmet_detection_amd.data import SyntheticVOC, TestAugmentor
from PIL import Image
ds = SyntheticVOC(transform=TestAugmentor(512), size=1, imsize=512, seed=3)
Image.fromarray(ds[0][0]).save('/tmp/img.jpg')
PY
timeout 240 python evaluate.py --data /tmp/img.jpg --model-load /tmp/w1/check_point_1.pth --imsize 512 --conf-th 0.3 --save-path /tmp/w1 > gpurun_out/fin8_demo.log 2>&1
echo "demo rc=$?"; tail -2 gpurun_out/fin8_demo.log; ls /tmp/w1/image.png
