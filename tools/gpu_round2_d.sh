#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests/test_gpu_parity.py tests/test_fg_gpu.py -m gpu -q > gpurun_out/pytest_gpu4.log 2>&1
echo "pytest rc=$?"
tail -3 gpurun_out/pytest_gpu4.log
timeout 300 python bench.py --steps 20 --warmup 3 --skip-cpu-baseline --skip-streaming --skip-config3 > gpurun_out/bench4.json 2> gpurun_out/bench4.log
echo "bench rc=$?"
python3 -c "import json; d=json.load(open('gpurun_out/bench4.json')); print('value', d['value'], 'ms_step', d['ms_per_step'], 'kernel_ms', d['roofline']['ms_per_launch'], 'frac', d['roofline']['frac'])"
