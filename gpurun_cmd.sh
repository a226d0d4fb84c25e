cd /root/repo
python -m pytest tests/test_gpu_gradcheck.py -x -q > gpurun_out/r2r_gradcheck.log 2>&1
echo "gradcheck rc=$?"; tail -3 gpurun_out/r2r_gradcheck.log | head -2
python -m pytest tests -m gpu -x -q > gpurun_out/r2r_pytest.log 2>&1
echo "full rc=$?"; tail -1 gpurun_out/r2r_pytest.log
