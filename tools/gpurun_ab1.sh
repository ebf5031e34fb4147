set -x
cd /root/repo
python -m pytest tests -m gpu -x -q 2>&1 | tail -5
for v in 0 2 1 9 6; do
  echo "=== VARIANT $v ==="
  DG_SCAN_VARIANT=$v timeout 300 python bench.py --steps 10 --warmup 3 --no-cpu-baseline --no-recall 2>/dev/null | python -c "import json,sys; j=json.load(sys.stdin); print('variant', '$v', 'qps', j['value'], 'ms', j['ms_per_step'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'], 'frac', j['roofline']['frac'])"
done
