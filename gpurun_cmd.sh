cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 330 python tools/quality_probe.py --steps 2000 --imgs 8 --size 256 > gpurun_out/quality_probe.log 2>&1
echo "QP_RC=$?"; grep -v Warning gpurun_out/quality_probe.log | tail -10
