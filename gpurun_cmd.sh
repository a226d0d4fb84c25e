cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 600 python tools/dbg_trace7.py > gpurun_out/dbg_trace7.log 2>&1
echo "DBG7_RC=$?"; grep -v "Warning\|warn\|amdgpu.ids" gpurun_out/dbg_trace7.log
