cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/fin6_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/fin6_pytest.log
python __graft_entry__.py smoke > gpurun_out/fin6_smoke.log 2>&1
echo "smoke rc=$?"
timeout 240 python bench.py --steps 30 --warmup 5 > gpurun_out/fin6_train.json 2>&1
echo train:; tail -1 gpurun_out/fin6_train.json
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph 2>/dev/null | tail -1 | python3 -c "import json,sys; print('b8:', json.loads(sys.stdin.read())['value'])"
timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph --fp8 2>/dev/null | tail -1 | python3 -c "import json,sys; print('fp8:', json.loads(sys.stdin.read())['value'])"
timeout 240 python bench.py --steps 20 --warmup 5 --num-stack 2 --increase-ch 128 2>/dev/null | tail -1 | python3 -c "import json,sys; print('big:', json.loads(sys.stdin.read())['value'])"
