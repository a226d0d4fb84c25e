cd /root/repo
python -m pytest tests/test_gpu_kernels.py -v -x > gpurun_out/dbg_full.log 2>&1
echo "rc=$?"; grep -E 'PASSED|FAILED|Fatal|Segmentation' gpurun_out/dbg_full.log | tail -8; tail -5 gpurun_out/dbg_full.log
