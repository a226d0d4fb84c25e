set -x
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
timeout 600 python tools/dbg_trace.py > gpurun_out/dbg_trace.log 2>&1
echo "DBG_RC=$?"; grep -v Warning gpurun_out/dbg_trace.log | tail -12
for C in 2048 4096 8192 12288; do
  RTHD_WGRAD_CHUNK=$C timeout 300 python tools/kbench.py wgrad --iters 30 > gpurun_out/kb_$C.log 2>&1
  echo "chunk=$C"; grep wgrad gpurun_out/kb_$C.log
done
timeout 300 python tools/kbench.py wgrad --iters 30 > gpurun_out/kb_def.log 2>&1
echo "chunk=default"; grep wgrad gpurun_out/kb_def.log
