set -x
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
rm -f jit_traced_model_*.pth
timeout 900 python -m pytest tests/test_gpu_kernels.py tests/test_gpu_e2e.py -x -q > gpurun_out/pytest_all.log 2>&1
echo "PYTEST_RC=$?"; tail -3 gpurun_out/pytest_all.log
timeout 600 python export.py --imsize 512 --save-path . > gpurun_out/export.log 2>&1
echo "EXPORT_RC=$?"; tail -2 gpurun_out/export.log
timeout 600 cmake -S tools/cpp_infer -B /tmp/cppb > gpurun_out/cpp_build.log 2>&1 \
  && timeout 600 cmake --build /tmp/cppb -j16 >> gpurun_out/cpp_build.log 2>&1
echo "CPPBUILD_RC=$?"
python - <<'PY'
from PIL import Image
import numpy as np
Image.fromarray((np.random.rand(512,512,3)*255).astype('uint8')).save('/tmp/img.ppm')
PY
KSO=$(ls real_time_helmet_detection_amd/ops/_C*.so | head -1)
timeout 300 /tmp/cppb/helmet_infer -m jit_traced_model_gpu.pth -i /tmp/img.ppm -n 500 -s 512 -k "$KSO" > gpurun_out/cpp_infer_gpu.log 2>&1
echo "CPPINFER_RC=$?"; head -2 gpurun_out/cpp_infer_gpu.log; tail -1 gpurun_out/cpp_infer_gpu.log
timeout 600 python bench.py --steps 20 --warmup 8 > gpurun_out/bench_train.json 2> gpurun_out/bench_train.log
echo "BT_RC=$?"; cat gpurun_out/bench_train.json
timeout 300 python tools/kbench.py bn --iters 30 > gpurun_out/kb_bn.log 2>&1
grep -v amdgpu gpurun_out/kb_bn.log
