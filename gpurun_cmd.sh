set -x
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
rm -f jit_traced_model_*.pth
timeout 1200 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "PYTEST_RC=$?"; tail -2 gpurun_out/pytest_gpu.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE-OK')" > gpurun_out/smoke.log 2>&1
echo "SMOKE_RC=$?"; tail -1 gpurun_out/smoke.log
timeout 600 python export.py --imsize 512 --save-path . > gpurun_out/export.log 2>&1
echo "EXPORT_RC=$?"
timeout 600 cmake -S tools/cpp_infer -B /tmp/cppb > gpurun_out/cpp_build.log 2>&1 && timeout 600 cmake --build /tmp/cppb -j16 >> gpurun_out/cpp_build.log 2>&1
echo "CPPBUILD_RC=$?"
python - <<'PY'
from PIL import Image
import numpy as np
Image.fromarray((np.random.rand(512,512,3)*255).astype('uint8')).save('/tmp/img.ppm')
PY
KSO=$(ls real_time_helmet_detection_amd/ops/_C*.so | head -1)
timeout 300 /tmp/cppb/helmet_infer -m jit_traced_model_gpu.pth -i /tmp/img.ppm -n 1000 -s 512 -k "$KSO" > gpurun_out/cpp_infer_gpu.log 2>&1
echo "CPP_RC=$?"; tail -1 gpurun_out/cpp_infer_gpu.log
timeout 600 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_train.json 2> gpurun_out/bench_train.log
echo BT=$?; cat gpurun_out/bench_train.json
timeout 600 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph > gpurun_out/bench_infer_b8.json 2> gpurun_out/bi8.log
echo BI8=$?; cat gpurun_out/bench_infer_b8.json
timeout 600 python bench.py --mode infer --batch-size 1 --steps 100 --warmup 20 --graph > gpurun_out/bench_infer_b1.json 2> gpurun_out/bi1.log
echo BI1=$?; cat gpurun_out/bench_infer_b1.json
