#!/bin/bash
# Round-2 profiling artifacts: kernel stats, HBM traffic (separate PMC
# passes per the gpurun rule), fused-vs-unfused A/B, throughput curve.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

# 1. kernel-trace stats of the default bench step
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r02 -o kt \
  -- python bench.py --steps 5 --warmup 2 --skip-cpu-baseline \
  --skip-streaming > gpurun_out/prof_kt.log 2>&1
echo "kt rc=$?"
python tools/rocpd_analyze.py gpurun_out/prof_r02/kt_results.db \
  > gpurun_out/rocprof_r02_kernel_stats.txt 2>&1
head -12 gpurun_out/rocprof_r02_kernel_stats.txt

# 2. HBM traffic: FETCH_SIZE and WRITE_SIZE in separate passes, 2^26
timeout 300 rocprofv3 --pmc FETCH_SIZE -d gpurun_out/prof_r02 -o fetch \
  -- python bench.py --steps 5 --warmup 2 --samples 67108864 \
  --skip-cpu-baseline --skip-streaming --skip-config3 \
  > gpurun_out/prof_fetch.log 2>&1
echo "fetch rc=$?"
timeout 300 rocprofv3 --pmc WRITE_SIZE -d gpurun_out/prof_r02 -o write \
  -- python bench.py --steps 5 --warmup 2 --samples 67108864 \
  --skip-cpu-baseline --skip-streaming --skip-config3 \
  > gpurun_out/prof_write.log 2>&1
echo "write rc=$?"
python tools/rocpd_analyze.py gpurun_out/prof_r02/fetch_results.db \
  > gpurun_out/traffic_fetch.txt 2>&1
python tools/rocpd_analyze.py gpurun_out/prof_r02/write_results.db \
  > gpurun_out/traffic_write.txt 2>&1
grep -i "decim4_fft" gpurun_out/traffic_fetch.txt gpurun_out/traffic_write.txt

# 3. fused vs unfused A/B
for env in "" "FSDR_CHAIN_FUSED=0" "FSDR_CHAIN_FFTFUSE=0"; do
  env $env timeout 180 python bench.py --steps 10 --warmup 2 \
    --skip-cpu-baseline --skip-streaming --skip-config3 2>/dev/null \
    | python3 -c "import json,sys; d=json.load(sys.stdin); print('[$env]', d['value'], 'GS/s-ish MS/s', d['ms_per_step'], 'ms')" \
    >> gpurun_out/fused_vs_unfused_r02.txt
done
cat gpurun_out/fused_vs_unfused_r02.txt

# 4. throughput vs batch size
for s in 16777216 67108864 268435456 1073741824; do
  timeout 240 python bench.py --steps 8 --warmup 2 --samples $s \
    --skip-cpu-baseline --skip-streaming --skip-config3 2>/dev/null \
    | python3 -c "import json,sys; d=json.load(sys.stdin); print('samples=$s', d['value'], 'MS/s', d['ms_per_step'], 'ms/step')" \
    >> gpurun_out/throughput_curve_r02.txt
done
cat gpurun_out/throughput_curve_r02.txt
