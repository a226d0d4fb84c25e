set -x
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
rm -f jit_traced_model_*.pth
timeout 600 python -m pytest tests/test_gpu_e2e.py -x -q -k "native or cache" > gpurun_out/pytest_nc.log 2>&1
echo "PY_RC=$?"; tail -2 gpurun_out/pytest_nc.log
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
timeout 300 /tmp/cppb/helmet_infer -m jit_traced_model_cpu.pth -i /tmp/img.ppm -n 5 -s 512 > gpurun_out/cpp_infer_cpu.log 2>&1
echo "CPPCPU_RC=$?"; tail -1 gpurun_out/cpp_infer_cpu.log
