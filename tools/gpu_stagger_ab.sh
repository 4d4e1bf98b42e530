#!/bin/bash
set -x
cd /root/repo
for sg in 0 2 4 8 12 16 24; do
  FSDR_CHAIN_STAGGER=$sg timeout 240 python bench.py --steps 10 --warmup 2 --skip-cpu-baseline --skip-streaming --skip-config3 2>/dev/null | python3 -c "import json,sys; d=json.load(sys.stdin); print('stagger=$sg', d['value'], d['roofline']['ms_per_launch'], d['roofline']['frac'])"
done
