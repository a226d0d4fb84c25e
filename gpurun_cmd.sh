cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/r2e_pytest.log 2>&1
echo "pytest rc=$?"; tail -2 gpurun_out/r2e_pytest.log
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph --fp8 > gpurun_out/r2e_infer_fp8.json 2>&1
echo fp8:; tail -1 gpurun_out/r2e_infer_fp8.json
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph > gpurun_out/r2e_infer_b8.json 2>&1
echo bf16:; tail -1 gpurun_out/r2e_infer_b8.json
timeout 240 python bench.py --steps 30 --warmup 5 > gpurun_out/r2e_bench1.json 2>&1
echo train:; tail -1 gpurun_out/r2e_bench1.json
timeout 120 python tools/kbench.py stem --iters 40 > gpurun_out/r2e_kbench_stem.log 2>&1
grep stem gpurun_out/r2e_kbench_stem.log
timeout 120 python tools/tr_probe.py > gpurun_out/r2e_trprobe.log 2>&1
echo "trprobe rc=$?"; head -20 gpurun_out/r2e_trprobe.log | tail -16
