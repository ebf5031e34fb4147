set -x
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 700 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_LDS_BANK_CONFLICT FETCH_SIZE -d gpurun_out/pmc_sq_c -o sq_c -- python bench.py --steps 3 --warmup 1 --no-cpu-baseline --no-recall > gpurun_out/pmc_sq_c.log 2>&1; echo CFGC_RC=$?
timeout 1100 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_LDS_BANK_CONFLICT FETCH_SIZE -d gpurun_out/pmc_sq_d -o sq_d -- python bench.py --kind ivf_pq --steps 2 --warmup 1 --n 100000000 --nlist 16384 --nprobe 64 --batch 4096 --no-cpu-baseline --no-recall > gpurun_out/pmc_sq_d.log 2>&1; echo CFGD_RC=$?
ls gpurun_out/pmc_sq_c gpurun_out/pmc_sq_d 2>/dev/null
