cd /root/repo
timeout 120 python tools/mfma_scale_probe.py > gpurun_out/r2f_msp.log 2>&1
echo "msp rc=$?"; tail -6 gpurun_out/r2f_msp.log
timeout 600 python tools/quality_probe.py holdout --steps 3000 --train-imgs 256 --val-imgs 64 --eval-every 500 > gpurun_out/r2f_holdout.log 2>&1
echo "holdout rc=$?"; grep -v libdrm gpurun_out/r2f_holdout.log | head -8
