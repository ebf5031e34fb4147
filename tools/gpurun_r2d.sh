set -x
cd /root/repo
timeout 1500 python -m pytest tests -m gpu -q -rA 2>&1 | tail -15
echo "=== cfg C (half-tile fast path) ==="
timeout 300 python bench.py --steps 15 --warmup 4 --no-cpu-baseline --no-recall 2>err.txt | tee gpurun_out/bench_r2_cfgC2.json | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'], 'frac', j['roofline']['frac'])" || tail -5 err.txt
echo "=== batch 8192 ==="
timeout 300 python bench.py --steps 5 --warmup 2 --batch 8192 --no-cpu-baseline --no-recall 2>err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'], 'frac', j['roofline']['frac'])" || tail -5 err.txt
echo "=== cfg D (PQ, new default) ==="
timeout 900 python bench.py --kind ivf_pq --steps 5 --warmup 2 --n 100000000 --nlist 16384 --nprobe 64 --batch 4096 --no-cpu-baseline --no-recall 2>err.txt | tee gpurun_out/bench_r2_cfgD.json | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'ms', j['ms_per_step'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'])" || tail -5 err.txt
