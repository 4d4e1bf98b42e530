#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
# quick numerics check of the occupancy variant + perf A/B
FSDR_CHAIN_OCC8=1 timeout 300 python -m pytest tests/test_gpu_parity.py -k "chain" -m gpu -q > gpurun_out/pytest_occ8.log 2>&1
echo "occ8 pytest rc=$?"; tail -2 gpurun_out/pytest_occ8.log
for env in "" "FSDR_CHAIN_OCC8=1"; do
  env $env timeout 180 python bench.py --steps 10 --warmup 2 --skip-cpu-baseline --skip-streaming --skip-config3 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print('[$env]', d['value'], d['roofline']['ms_per_launch'], d['roofline']['frac'])"
done
