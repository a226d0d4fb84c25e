cd /root/repo
python -m pytest tests/test_gpu_kernels.py -x -q 2>&1 | tail -1
timeout 180 python bench.py --mode infer --batch-size 1 --steps 100 --warmup 20 --graph 2>/dev/null | tail -1 | python3 -c "import json,sys; print('b1:', json.loads(sys.stdin.read())['value'])"
