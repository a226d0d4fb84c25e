cd /root/repo
for i in 1 2 3; do
  timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph 2>/dev/null | tail -1 | python3 -c "import json,sys; print('bf16:', json.loads(sys.stdin.read())['value'])"
  timeout 180 python bench.py --mode infer --batch-size 8 --steps 50 --warmup 10 --graph --fp8 2>/dev/null | tail -1 | python3 -c "import json,sys; print('fp8 :', json.loads(sys.stdin.read())['value'])"
done
