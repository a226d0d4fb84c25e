cd /tmp
export TMPDIR=/tmp
mkdir -p /root/repo/gpurun_out
timeout 240 rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_INSTS_MFMA SQ_INSTS_VALU SQ_WAIT_INST_ANY SQ_WAVE_CYCLES SQ_ACTIVE_INST_ANY -d /tmp/pmcw2 -- python /root/repo/tools/kbench.py wgrad --iters 10 > /root/repo/gpurun_out/pmc2.log 2>&1
echo "PMC_RC=$?"
cp -r /tmp/pmcw2 /root/repo/gpurun_out/pmcw2 2>/dev/null
cd /root/repo
timeout 240 python tools/kbench.py conv --iters 30 2>/dev/null | grep -i conv
