cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/r2u_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/r2u_pytest.log
for i in 1 2 3; do
  timeout 240 python bench.py --steps 30 --warmup 5 2>/dev/null | tail -1 | python -c "import json,sys; d=json.loads(sys.stdin.read()); print('train:', d['value'], d['ms_per_step'])"
done
