cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof_b1v2 -o p -- python /root/repo/bench.py --mode infer --batch-size 1 --steps 60 --warmup 10 > /root/repo/gpurun_out/r2v_b1prof.log 2>&1
echo "b1 prof rc=$?"
cd /root/repo
timeout 900 python tools/quality_probe.py holdout --steps 2000 --train-imgs 192 --val-imgs 48 --eval-every 400 --in-ch 128 --size 512 --batch-size 8 --lr 1e-3 > gpurun_out/r2v_holdout512.log 2>&1
echo "holdout512 rc=$?"; grep -v libdrm gpurun_out/r2v_holdout512.log | head -6
