cd /root/repo
timeout 180 python tools/kbench.py fp8 --iters 50 > gpurun_out/r2h_kbench_fp8.log 2>&1
grep conv gpurun_out/r2h_kbench_fp8.log
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_ACTIVE_INST_ANY,SQ_WAIT_INST_ANY,SQ_INSTS_MFMA,SQ_INSTS_VALU,SQ_LDS_BANK_CONFLICT --output-format csv -d /root/repo/gpurun_out/pmc_fp8 -o pmc -- python /root/repo/tools/kbench.py fp8 --iters 10 > /root/repo/gpurun_out/r2h_pmc.log 2>&1
echo "pmc rc=$?"; ls /root/repo/gpurun_out/pmc_fp8/ 2>/dev/null | head -3
