set -x
cd /root/repo
for v in 12 10; do
  echo "=== VARIANT $v quick parity ==="
  DG_SCAN_VARIANT=$v timeout 300 python -m pytest tests/test_gpu_parity.py -x -q -m gpu -k "ivf_parity or ivf_recall" 2>&1 | tail -2
  echo "=== VARIANT $v cfg C bench ==="
  DG_SCAN_VARIANT=$v timeout 300 python bench.py --steps 10 --warmup 3 --no-cpu-baseline --no-recall 2>bench_err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'], 'frac', j['roofline']['frac'])" || tail -4 bench_err.txt
done
