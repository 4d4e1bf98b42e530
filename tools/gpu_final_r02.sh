#!/bin/bash
# Round-2 final validation: full GPU suite, extended chain fuzz,
# single-rank torchrun RCCL re-check, canonical default bench line.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_final.log 2>&1
echo "pytest rc=$?"
tail -3 gpurun_out/pytest_final.log
timeout 600 python tools/fuzz_chain.py 60 > gpurun_out/fuzz_final.log 2>&1
echo "fuzz rc=$?"
tail -4 gpurun_out/fuzz_final.log
timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
  --master-addr 127.0.0.1 --master-port 29533 bench.py --gpus 1 --steps 8 \
  --warmup 2 --skip-cpu-baseline --skip-streaming --skip-config3 \
  > gpurun_out/bench_tr_final.json 2> gpurun_out/bench_tr_final.log
echo "torchrun rc=$?"
tail -1 gpurun_out/bench_tr_final.json
timeout 420 python bench.py --steps 20 --warmup 3 \
  > gpurun_out/bench_final.json 2> gpurun_out/bench_final.log
echo "bench rc=$?"
tail -1 gpurun_out/bench_final.json
