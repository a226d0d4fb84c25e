cd /root/repo
timeout 300 python main.py --train-flag --synthetic --synthetic-size 128 --batch-size 16 --amp --end-epoch 1 --print-interval 8 --num-workers 4 --save-path /tmp/w1 > /dev/null 2>&1
timeout 300 python export.py --model-load /tmp/w1/check_point_1.pth --num-stack 1 --hourglass-inch 128 --save-path /tmp/w1 > gpurun_out/fin7_export.log 2>&1
echo "export rc=$?"
cmake -S tools/cpp_infer -B /tmp/cppb -DTORCH_ROOT=$(python -c 'import torch, os; print(os.path.dirname(torch.__file__))') > /dev/null 2>&1 && cmake --build /tmp/cppb -j 16 > /dev/null 2>&1
python3 - <<'PY'
from real_time_helmet_detection_amd.data import SyntheticVOC, TestAugmentor
ds = SyntheticVOC(transform=TestAugmentor(512), size=1, imsize=512, seed=3)
img = ds[0][0]
with open('/tmp/img.ppm','wb') as f:
    f.write(b'P6\n512 512\n255\n'); f.write(img.tobytes())
from PIL import Image
Image.fromarray(img).save('/tmp/img.jpg')
PY
timeout 240 /tmp/cppb/helmet_infer -m /tmp/w1/jit_traced_model_gpu.pth -i /tmp/img.ppm -k real_time_helmet_detection_amd/ops/_C.cpython-310-x86_64-linux-gnu.so -b > gpurun_out/fin7_cpp.log 2>&1
echo "cpp rc=$?"; tail -1 gpurun_out/fin7_cpp.log
# GPU demo path (reference evaluate.py __main__)
timeout 240 python evaluate.py --data /tmp/img.jpg --model-load /tmp/w1/check_point_1.pth --imsize 512 --conf-th 0.3 --save-path /tmp/w1 > gpurun_out/fin7_demo.log 2>&1
echo "demo rc=$?"; tail -2 gpurun_out/fin7_demo.log
