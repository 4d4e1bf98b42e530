set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest rc=$?"
tail -5 gpurun_out/pytest_gpu.log
timeout 420 python bench.py --steps 20 --warmup 3 > gpurun_out/bench_default.json 2> gpurun_out/bench_default.log
echo "bench rc=$?"
tail -2 gpurun_out/bench_default.json
timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 --master-addr 127.0.0.1 --master-port 29512 bench.py --gpus 1 --steps 10 --warmup 2 --skip-cpu-baseline --skip-streaming > gpurun_out/bench_tr1.json 2> gpurun_out/bench_tr1.log
echo "torchrun1 rc=$?"
tail -2 gpurun_out/bench_tr1.json
timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29513 bench.py --gpus 2 --steps 10 --warmup 2 --skip-cpu-baseline --skip-streaming --samples 67108864 > gpurun_out/bench_tr2.json 2> gpurun_out/bench_tr2.log
echo "torchrun2 rc=$?"
tail -2 gpurun_out/bench_tr2.json
tail -15 gpurun_out/bench_tr2.log
