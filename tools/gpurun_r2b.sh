set -x
cd /root/repo
timeout 1500 python -m pytest tests -m gpu -q 2>&1 | tail -4
echo "=== nq=1 latency (hipGraph replay) ==="
timeout 300 python bench.py --batch 1 --steps 300 --warmup 30 --no-cpu-baseline --no-recall 2>err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'ms_per_call', j['ms_per_step'])" || tail -6 err.txt
echo "=== nq=1 latency without graph ==="
DG_NO_GRAPH=1 timeout 300 python bench.py --batch 1 --steps 300 --warmup 30 --no-cpu-baseline --no-recall 2>err.txt | python -c "import json,sys; j=json.load(sys.stdin); print('qps', j['value'], 'ms_per_call', j['ms_per_step'])" || tail -6 err.txt
echo "=== PQ occupancy A/B (cfg D) ==="
timeout 1200 python tools/pq_ab.py 2>pq_err.txt || tail -8 pq_err.txt
echo "=== PMC FETCH_SIZE (cfg C scan traffic) ==="
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 900 rocprofv3 --pmc FETCH_SIZE -d gpurun_out/pmc_r2 -o pmc_r2 -- python bench.py --steps 3 --warmup 1 --no-cpu-baseline --no-recall > gpurun_out/pmc_r2_bench.log 2>&1 && echo PMC_OK || tail -3 gpurun_out/pmc_r2_bench.log
find gpurun_out/pmc_r2* -name "*.csv" 2>/dev/null | head -3
