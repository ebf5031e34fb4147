set -x
cd /root/repo
timeout 1200 python -m pytest tests -m gpu -q > gpurun_out/pytest_r2f.log 2>&1
echo PYTEST_RC=$?
tail -3 gpurun_out/pytest_r2f.log
echo "=== PQ select+COOP A/B (cfg D) ==="
timeout 1200 python tools/pq_ab_c.py 2>pq_err.txt || tail -8 pq_err.txt
echo "=== recall curve (cfg C) ==="
timeout 600 python bench.py --steps 10 --warmup 3 --no-cpu-baseline --recall-curve 2>curve_err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'])" || tail -4 curve_err.txt
tail -c 800 gpurun_out/recall_curve.json 2>/dev/null
