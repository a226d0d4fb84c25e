cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/fin9_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/fin9_pytest.log
python __graft_entry__.py smoke 2>&1 | tail -1
timeout 240 python bench.py 2>/dev/null | tail -1 | python3 -c "import json,sys; d=json.loads(sys.stdin.read()); print('default bench:', d['value'], 'img/s,', d['ms_per_step'], 'ms/step')"
