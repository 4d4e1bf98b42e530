#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu3.log 2>&1
echo "pytest rc=$?"
tail -4 gpurun_out/pytest_gpu3.log
timeout 300 python tools/resamp_ab.py > gpurun_out/resamp_ab3.log 2>&1
echo "resamp rc=$?"
cat gpurun_out/resamp_ab3.log
timeout 420 python bench.py --steps 20 --warmup 3 --skip-cpu-baseline --skip-streaming > gpurun_out/bench3.json 2> gpurun_out/bench3.log
echo "bench rc=$?"
tail -1 gpurun_out/bench3.json
