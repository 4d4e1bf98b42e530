#!/bin/bash
# Round-2 GPU batch B: generalized fused kernel + WLAN e2e + tiled
# resampler A/B + config-3 leg + kernel stats.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu2.log 2>&1
echo "pytest rc=$?"
tail -4 gpurun_out/pytest_gpu2.log

timeout 300 python tools/resamp_ab.py > gpurun_out/resamp_ab.log 2>&1
echo "resamp rc=$?"
cat gpurun_out/resamp_ab.log

timeout 420 python bench.py --steps 20 --warmup 3 --skip-cpu-baseline \
    > gpurun_out/bench2.json 2> gpurun_out/bench2.log
echo "bench rc=$?"
tail -1 gpurun_out/bench2.json

export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof2 -o r02a \
    -- python bench.py --steps 5 --warmup 2 --skip-cpu-baseline \
    --skip-streaming --skip-config3 > gpurun_out/prof2.log 2>&1
echo "rocprof rc=$?"
for f in $(find gpurun_out/prof2 -name "*kernel_stats*"); do
  echo "== $f"; head -15 "$f"
done
