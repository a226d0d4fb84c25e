cd /root/repo
python -m pytest tests -m gpu -x -q > gpurun_out/fin3_pytest.log 2>&1
echo "pytest rc=$?"; tail -1 gpurun_out/fin3_pytest.log
timeout 700 python tools/bench_all.py --out gpurun_out/fin3_bench_all.json > /dev/null 2>&1
echo "bench_all rc=$?"
python3 -c "
import json
r = json.load(open('gpurun_out/fin3_bench_all.json'))
for k, v in r['configs'].items():
    res = v.get('result')
    print(k, '->', (res['value'], res['metric']) if res else v.get('ok', v.get('rc')))
"
