cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/fin4_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/fin4_pytest.log
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof_final -o p -- python /root/repo/bench.py --steps 15 --warmup 4 --no-train-graph > /root/repo/gpurun_out/fin4_prof.log 2>&1
echo "prof rc=$?"
timeout 240 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_ACTIVE_INST_ANY,SQ_WAIT_INST_ANY,SQ_INSTS_MFMA,SQ_INSTS_VALU,SQ_LDS_BANK_CONFLICT --output-format csv -d /root/repo/gpurun_out/pmc_final -o p -- python /root/repo/tools/kbench.py conv --iters 8 > /root/repo/gpurun_out/fin4_pmc.log 2>&1
echo "pmc rc=$?"
