#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_fin -o kt -- \
  python bench.py --steps 5 --warmup 2 --skip-cpu-baseline --skip-streaming \
  > gpurun_out/kt_fin.log 2>&1
echo "kt rc=$?"
python tools/rocpd_analyze.py gpurun_out/prof_fin/kt_results.db > gpurun_out/rocprof_r02_final.txt 2>&1
head -14 gpurun_out/rocprof_r02_final.txt
timeout 300 rocprofv3 --pmc FETCH_SIZE -d gpurun_out/prof_fin -o fetch -- \
  python bench.py --steps 6 --warmup 2 --samples 67108864 --skip-cpu-baseline \
  --skip-streaming --skip-config3 --skip-roofline > gpurun_out/fetch_fin.log 2>&1
echo "fetch rc=$?"
timeout 300 rocprofv3 --pmc WRITE_SIZE -d gpurun_out/prof_fin -o write -- \
  python bench.py --steps 6 --warmup 2 --samples 67108864 --skip-cpu-baseline \
  --skip-streaming --skip-config3 --skip-roofline > gpurun_out/write_fin.log 2>&1
echo "write rc=$?"
python tools/rocpd_analyze.py gpurun_out/prof_fin/fetch_results.db 2>&1 | grep -E "PMC.*ap_tpl|PMC.*fill"
python tools/rocpd_analyze.py gpurun_out/prof_fin/write_results.db 2>&1 | grep -E "PMC.*ap_tpl|PMC.*fill"
