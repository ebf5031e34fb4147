set -x
cd /root/repo
echo "=== new/changed tests, verbose, no -x ==="
timeout 900 python -m pytest tests/test_gpu_parity.py -q -m gpu -k "wide or odd or large_dim or graph or error_paths or filter or shard" -rA 2>&1 | tail -25
echo "=== core parity subset ==="
timeout 600 python -m pytest tests/test_gpu_parity.py -q -m gpu -k "flat_parity or ivf_parity or ivfpq_parity or train" 2>&1 | tail -3
echo "=== batch 8192 with QTM=24 dense tile ==="
timeout 300 python bench.py --steps 5 --warmup 2 --batch 8192 --no-cpu-baseline --no-recall 2>err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'], 'frac', j['roofline']['frac'])" || tail -5 err.txt
echo "=== cfg C regression ==="
timeout 300 python bench.py --steps 10 --warmup 3 --no-cpu-baseline --no-recall 2>err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'], 'frac', j['roofline']['frac'])" || tail -5 err.txt
