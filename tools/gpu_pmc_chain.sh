#!/bin/bash
# PMC wait/active breakdown of the fused chain kernel + quick env A/Bs.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
timeout 420 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY SQ_ACTIVE_INST_ANY SQ_WAIT_INST_LDS -d gpurun_out/pmc_r02 -o sq -- \
  python bench.py --steps 5 --warmup 2 --skip-cpu-baseline --skip-streaming --skip-config3 --samples 67108864 > gpurun_out/pmc_sq.log 2>&1
echo "pmc rc=$?"
python tools/rocpd_analyze.py gpurun_out/pmc_r02/*.db > gpurun_out/pmc_sq_summary.txt 2>&1
head -40 gpurun_out/pmc_sq_summary.txt
for cap in 4096 8192 16384 32768; do
  FSDR_FIR_GRID_CAP=$cap timeout 180 python bench.py --steps 8 --warmup 2 --skip-cpu-baseline --skip-streaming --skip-config3 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print('cap=$cap', d['value'], d['roofline']['ms_per_launch'])"
done
FSDR_CHAIN_BLOCK512=1 timeout 180 python bench.py --steps 8 --warmup 2 --skip-cpu-baseline --skip-streaming --skip-config3 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print('block512', d['value'], d['roofline']['ms_per_launch'])"
