set -x
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
# 1) full GPU test suite
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "PYTEST_RC=$?" | tee -a gpurun_out/pytest_gpu.log
tail -5 gpurun_out/pytest_gpu.log
# 2) export traced models (gpu+cpu) from a random-init checkpoint-shaped net
timeout 600 python export.py --imsize 512 > gpurun_out/export.log 2>&1
echo "EXPORT_RC=$?" | tee -a gpurun_out/export.log
ls -la jit_traced_model_*.pth >> gpurun_out/export.log 2>&1
# 3) build C++ app and run on the GPU traced model
timeout 600 cmake -S tools/cpp_infer -B /tmp/cppb > gpurun_out/cpp_build.log 2>&1 \
  && timeout 600 cmake --build /tmp/cppb -j16 >> gpurun_out/cpp_build.log 2>&1
echo "CPPBUILD_RC=$?" | tee -a gpurun_out/cpp_build.log
python - <<'PY'
from PIL import Image
import numpy as np
Image.fromarray((np.random.rand(512,512,3)*255).astype('uint8')).save('/tmp/img.ppm')
PY
timeout 300 /tmp/cppb/helmet_infer -m jit_traced_model_gpu.pth -i /tmp/img.ppm -n 200 -s 512 > gpurun_out/cpp_infer.log 2>&1
echo "CPPINFER_RC=$?" | tee -a gpurun_out/cpp_infer.log
tail -3 gpurun_out/cpp_infer.log
# 4) final bench numbers: train + infer b8 + infer b1
timeout 600 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_train.json 2> gpurun_out/bench_train.log
echo "BTRAIN_RC=$?"
timeout 600 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph > gpurun_out/bench_infer_b8.json 2> gpurun_out/bench_infer_b8.log
echo "BINFER8_RC=$?"
timeout 600 python bench.py --mode infer --batch-size 1 --steps 100 --warmup 20 --graph > gpurun_out/bench_infer_b1.json 2> gpurun_out/bench_infer_b1.log
echo "BINFER1_RC=$?"
tail -1 gpurun_out/bench_train.json gpurun_out/bench_infer_b8.json gpurun_out/bench_infer_b1.json
