cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/fin2_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/fin2_pytest.log
python __graft_entry__.py smoke > gpurun_out/fin2_smoke.log 2>&1
echo "smoke rc=$?"; tail -1 gpurun_out/fin2_smoke.log
for i in 1 2; do
  timeout 240 python bench.py --steps 30 --warmup 5 2>/dev/null | tail -1 | python3 -c "import json,sys; print('train:', json.loads(sys.stdin.read())['value'])"
done
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph 2>/dev/null | tail -1 | python3 -c "import json,sys; print('b8:', json.loads(sys.stdin.read())['value'])"
