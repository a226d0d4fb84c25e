cd /root/repo
for i in 1 2; do
  timeout 240 python bench.py --steps 30 --warmup 5 2>/dev/null | tail -1 | python -c "import json,sys; print('fused  :', json.loads(sys.stdin.read())['value'])"
  RTHD_NO_FUSED_STATS=1 timeout 240 python bench.py --steps 30 --warmup 5 2>/dev/null | tail -1 | python -c "import json,sys; print('nofused:', json.loads(sys.stdin.read())['value'])"
done
