set -x
cd /root/repo
timeout 1200 python -m pytest tests -m gpu -x -q 2>&1 | tail -4
echo "=== default (v12) cfg C with recall + cross-engine ==="
timeout 600 python bench.py --steps 20 --warmup 5 2>err.txt | tee gpurun_out/bench_r2_cfgC.json | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'], 'frac', j['roofline']['frac'], 'recall', j['recall_at_k'], 'xeng', j.get('recall_cross_engine'))" || tail -6 err.txt
echo "=== batch 8192 (v12) ==="
timeout 300 python bench.py --steps 5 --warmup 2 --batch 8192 --no-cpu-baseline --no-recall 2>err.txt | tee gpurun_out/bench_r2_cfgE1.json | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'scan_ms', j['roofline']['detail']['scan_ms_per_launch'], 'frac', j['roofline']['frac'])" || tail -4 err.txt
echo "=== Flat cfg B ==="
timeout 300 python bench.py --kind flat --n 1000000 --batch 256 --steps 10 --warmup 3 --no-cpu-baseline 2>err.txt | tee gpurun_out/bench_r2_cfgB.json | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'frac', j['roofline']['frac'])" || tail -4 err.txt
echo "=== PMC traffic for the default scan (cfg C) ==="
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE -d gpurun_out/pmc_r2 -o pmc_r2 -- python bench.py --steps 3 --warmup 1 --no-cpu-baseline --no-recall > gpurun_out/pmc_r2_bench.log 2>&1 || tail -4 gpurun_out/pmc_r2_bench.log
ls gpurun_out/pmc_r2* 2>/dev/null | head
