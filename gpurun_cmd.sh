cd /tmp && export TMPDIR=/tmp
cd /root/repo
timeout 300 bash -c 'cd /tmp && rocprofv3 --kernel-trace --stats -- python /root/repo/bench.py --steps 5 --warmup 2 --no-train-graph' > gpurun_out/r2c_prof.log 2>&1
echo "prof rc=$?"
grep -A 40 'KERNEL_NAME\|NAME' gpurun_out/r2c_prof.log | head -50
for ch in 512 1024 2048 4096; do
  echo "== RTHD_WGRAD_CHUNK=$ch =="
  RTHD_WGRAD_CHUNK=$ch timeout 120 python tools/kbench.py wgrad --iters 40 2>/dev/null | grep wgrad
done > gpurun_out/r2c_wgrad_sweep.log 2>&1
echo "sweep rc=$?"; cat gpurun_out/r2c_wgrad_sweep.log
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph > gpurun_out/r2c_infer_b8.json 2>&1
tail -1 gpurun_out/r2c_infer_b8.json
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph --fp8 > gpurun_out/r2c_infer_fp8.json 2>&1
tail -1 gpurun_out/r2c_infer_fp8.json
