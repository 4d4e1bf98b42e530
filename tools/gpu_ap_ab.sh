#!/bin/bash
set -x
cd /root/repo
FSDR_CHAIN_ALLPHASE=1 timeout 300 python -m pytest tests/test_gpu_parity.py -k "chain" -m gpu -q 2>&1 | tail -2
for env in "" "FSDR_CHAIN_ALLPHASE=1"; do
  env $env timeout 240 python bench.py --steps 10 --warmup 2 --skip-cpu-baseline --skip-streaming --skip-config3 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print('[$env]', d['value'], d['roofline']['ms_per_launch'], d['roofline']['frac'])"
done
