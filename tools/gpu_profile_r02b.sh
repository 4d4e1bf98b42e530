#!/bin/bash
# Clean r02 traffic passes (no roofline pollution) + kernel stats at the
# new 2^30 default + full test suite + the reference bench line.
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu5.log 2>&1
echo "pytest rc=$?"
tail -3 gpurun_out/pytest_gpu5.log

timeout 300 rocprofv3 --pmc FETCH_SIZE -d gpurun_out/prof_r02b -o fetch -- \
  python bench.py --steps 6 --warmup 2 --samples 67108864 \
  --skip-cpu-baseline --skip-streaming --skip-config3 --skip-roofline \
  > gpurun_out/fetch2.log 2>&1
echo "fetch rc=$?"
timeout 300 rocprofv3 --pmc WRITE_SIZE -d gpurun_out/prof_r02b -o write -- \
  python bench.py --steps 6 --warmup 2 --samples 67108864 \
  --skip-cpu-baseline --skip-streaming --skip-config3 --skip-roofline \
  > gpurun_out/write2.log 2>&1
echo "write rc=$?"
python tools/rocpd_analyze.py gpurun_out/prof_r02b/fetch_results.db 2>&1 | grep -E "decim4_fft|fill"
python tools/rocpd_analyze.py gpurun_out/prof_r02b/write_results.db 2>&1 | grep -E "decim4_fft|fill"

timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r02b -o kt30 \
  -- python bench.py --steps 5 --warmup 2 --skip-cpu-baseline \
  --skip-streaming > gpurun_out/kt30.log 2>&1
echo "kt30 rc=$?"
python tools/rocpd_analyze.py gpurun_out/prof_r02b/kt30_results.db \
  > gpurun_out/rocprof_r02_kernel_stats.txt 2>&1
head -14 gpurun_out/rocprof_r02_kernel_stats.txt

timeout 420 python bench.py --steps 20 --warmup 3 > gpurun_out/bench_r02.json 2> gpurun_out/bench_r02.log
echo "bench rc=$?"
tail -1 gpurun_out/bench_r02.json
