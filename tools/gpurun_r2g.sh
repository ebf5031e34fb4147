set -x
cd /root/repo
echo "=== mirror selftest + core parity ==="
timeout 600 python -m pytest tests/test_gpu_parity.py -q -m gpu -k "cpp_mirror or flat_parity or ivf_parity" 2>&1 | tail -3
echo "=== cfg B: hand MFMA forced ==="
DG_GEMM=mfma timeout 300 python bench.py --kind flat --n 1000000 --batch 256 --steps 10 --warmup 3 --no-cpu-baseline 2>err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'ms', j['ms_per_step'], 'frac', j['roofline']['frac'])" || tail -5 err.txt
echo "=== cfg B: rocBLAS (default at this shape) ==="
timeout 300 python bench.py --kind flat --n 1000000 --batch 256 --steps 10 --warmup 3 --no-cpu-baseline 2>err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'ms', j['ms_per_step'], 'frac', j['roofline']['frac'])" || tail -5 err.txt
echo "=== cfg C regression (GEMM change touches coarse) ==="
timeout 300 python bench.py --steps 10 --warmup 3 --no-cpu-baseline --no-recall 2>err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'], 'coarse_ms', j['roofline']['detail']['coarse_ms'])" || tail -5 err.txt
