cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
rm -f jit_traced_model_*.pth
timeout 300 python export.py --imsize 512 --save-path . > gpurun_out/export.log 2>&1
echo "EXPORT_RC=$?"; tail -1 gpurun_out/export.log
timeout 400 cmake -S tools/cpp_infer -B /tmp/cppb > gpurun_out/cpp_build.log 2>&1 && timeout 400 cmake --build /tmp/cppb -j16 >> gpurun_out/cpp_build.log 2>&1
echo "CPPBUILD_RC=$?"
python - <<'PY'
from PIL import Image
import numpy as np
Image.fromarray((np.random.rand(512,512,3)*255).astype('uint8')).save('/tmp/img.ppm')
PY
KSO=$(ls real_time_helmet_detection_amd/ops/_C*.so | head -1)
timeout 200 /tmp/cppb/helmet_infer -m jit_traced_model_gpu.pth -i /tmp/img.ppm -n 1000 -s 512 -k "$KSO" > gpurun_out/cpp_infer_gpu.log 2>&1
echo "CPP_RC=$?"; tail -1 gpurun_out/cpp_infer_gpu.log
