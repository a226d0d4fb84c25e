cd /root/repo
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph --fp8 > gpurun_out/r2y_fp8.json 2>&1
echo fp8:; tail -1 gpurun_out/r2y_fp8.json
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph > gpurun_out/r2y_b8.json 2>&1
echo bf16:; tail -1 gpurun_out/r2y_b8.json
python -m pytest tests/test_gpu_e2e.py -x -q > gpurun_out/r2y_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/r2y_pytest.log
# C++ app with the faster NMS/decode
timeout 300 python main.py --train-flag --synthetic --synthetic-size 128 --batch-size 16 --amp --end-epoch 1 --print-interval 8 --num-workers 4 --save-path /tmp/w1 > /dev/null 2>&1
timeout 300 python export.py --model-load /tmp/w1/check_point_1.pth --num-stack 1 --hourglass-inch 128 --save-path /tmp/w1 > /dev/null 2>&1
cmake -S tools/cpp_infer -B /tmp/cppb -DTORCH_ROOT=$(python -c 'import torch, os; print(os.path.dirname(torch.__file__))') > /dev/null 2>&1 && cmake --build /tmp/cppb -j 16 > /dev/null 2>&1
python - <<'PY'
from real_time_helmet_detection_amd.data import SyntheticVOC, TestAugmentor
ds = SyntheticVOC(transform=TestAugmentor(512), size=1, imsize=512, seed=3)
img = ds[0][0]
with open('/tmp/img.ppm','wb') as f:
    f.write(b'P6\n512 512\n255\n'); f.write(img.tobytes())
PY
timeout 240 /tmp/cppb/helmet_infer -m /tmp/w1/jit_traced_model_gpu.pth -i /tmp/img.ppm -k real_time_helmet_detection_amd/ops/_C.cpython-310-x86_64-linux-gnu.so -b > gpurun_out/r2y_cpp.log 2>&1
echo "cpp rc=$?"; tail -1 gpurun_out/r2y_cpp.log
