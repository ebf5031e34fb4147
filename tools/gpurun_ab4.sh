set -x
cd /root/repo
DG_SCAN_VARIANT=10 timeout 900 python -m pytest tests/test_gpu_parity.py -x -q -m gpu 2>&1 | tail -3
for v in 10 11 0; do
  echo "=== VARIANT $v cfg C ==="
  DG_SCAN_VARIANT=$v timeout 300 python bench.py --steps 10 --warmup 3 --no-cpu-baseline --no-recall 2>bench_err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'ms', j['ms_per_step'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'], 'frac', j['roofline']['frac'])" || tail -6 bench_err.txt
done
echo "=== VARIANT 10 batch 8192 ==="
DG_SCAN_VARIANT=10 timeout 300 python bench.py --steps 5 --warmup 2 --batch 8192 --no-cpu-baseline --no-recall 2>bench_err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'ms', j['ms_per_step'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'], 'frac', j['roofline']['frac'])" || tail -6 bench_err.txt
echo "=== Flat cfg B ==="
timeout 300 python bench.py --kind flat --n 1000000 --batch 256 --steps 10 --warmup 3 --no-cpu-baseline 2>bench_err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'ms', j['ms_per_step'], 'frac', j['roofline']['frac'])" || tail -6 bench_err.txt
