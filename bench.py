#!/usr/bin/env python3
"""bench.py — BASELINE.json metric: MSample/s of Complex32 input through
the 127-tap FIR -> decim-4 -> 1024-pt FFT chain, on N GPUs of one node.

A step = one pass of the chain over one HBM-resident batch of synthetic
samples (re/im iid uniform[-1,1), device-generated, seeded). At N>1 each
rank runs an independent channel on its own GPU (config 4) and the
spectrum-combine step joins the ranks' spectra with one RCCL all-gather
per step. The payload is the reference spectrum SINK's output — the
MovingAvg-averaged 1024-bin magnitude spectrum (moving_avg.rs is what the
spectrum app displays), one averaged frame per rank per step (4 KiB) —
computed on-GPU from every frame's |X|^2. Gathering every RAW frame
instead (= benchmarking the interconnect, ~67 MB/rank/step) is available
with --gather-frames. Weak scaling. Launch for N>1:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N ...

Rank 0 prints ONE JSON line (the driver contract).
"""
import argparse
import ctypes
import json
import os
import sys
import time

import numpy as np

# Import torch (and its bundled HIP runtime) BEFORE the futuresdr_amd C-ABI
# library: loading /opt/rocm's libamdhip64 first makes torch's lazy CUDA/HIP
# init fail with "No HIP GPUs are available" (observed on the GPU box).
import torch  # noqa: E402

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

FP32_PEAK_TFLOPS = 157.3  # gfx950 fp32 vector/MFMA peak (MI355X_MICROARCH.md)


def log(msg):
    print(msg, file=sys.stderr, flush=True)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--samples", type=int, default=1 << 28,
                   help="chain-input Complex32 samples per step per GPU")
    p.add_argument("--fft", type=int, default=1024)
    p.add_argument("--decim", type=int, default=4)
    p.add_argument("--taps", type=int, default=127)
    p.add_argument("--gather-frames", action="store_true",
                   help="all-gather every raw magnitude frame instead of "
                        "the MovingAvg-averaged spectrum")
    p.add_argument("--skip-cpu-baseline", action="store_true")
    p.add_argument("--cpu-sample", type=int, default=0,
                   help="fixed CPU-baseline sample size (0 = auto ~10s)")
    p.add_argument("--traffic-file", default=os.path.join(
        REPO, "profiles", "traffic_r01.json"))
    return p.parse_args()


def alloc_dev(lib, bytes_):
    p = ctypes.c_void_p()
    if lib.fsdr_dev_alloc(ctypes.byref(p), bytes_) != 0:
        raise RuntimeError(lib.fsdr_last_error().decode())
    return p


def measure_chain_roofline(fa, torch, chain, d_in, n_samples, taps1,
                           taps2, decim, fft_len, traffic_file):
    """Dominant-kernel roofline. The default chain is ONE kernel
    (k_decim4_fft_mfma_tpl<80>: the algebraically-fused 253-tap
    decimating filter with an in-block 1024-pt FFT — DESIGN.md §d), so
    the launch IS the dominant kernel; duration via HIP events on the
    launch stream. achieved = algorithmic flops per launch: fused-filter
    (T1+T2-1 taps x 4 flops per decimated output) + FFT (5 N log2 N per
    frame), all fp32."""
    st = torch.cuda.current_stream()
    g_len = taps1.size + taps2.size - 1
    produced = (n_samples + 1 - g_len) // decim
    frames = produced // fft_len
    prod = frames * fft_len
    for _ in range(3):
        chain.run_dev(d_in.value, n_samples, 0, 0, 0, 0,
                      stream=st.cuda_stream)
    torch.cuda.synchronize()
    reps = 20
    ev0 = torch.cuda.Event(enable_timing=True)
    ev1 = torch.cuda.Event(enable_timing=True)
    ev0.record(st)
    for _ in range(reps):
        chain.run_dev(d_in.value, n_samples, 0, 0, 0, 0,
                      stream=st.cuda_stream)
    ev1.record(st)
    torch.cuda.synchronize()
    ms = ev0.elapsed_time(ev1) / reps
    import math
    flops = prod * g_len * 4 + frames * 5 * fft_len * math.log2(fft_len)
    achieved_tf = flops / (ms * 1e-3) / 1e12
    traffic = None
    if traffic_file and os.path.exists(traffic_file):
        with open(traffic_file) as f:
            t = json.load(f)
        bps = t.get("chain_kernel_hbm_bytes_per_input_sample")
        if bps is not None:
            traffic = bps * n_samples
    return {
        "bound": "mfma",  # fp32 MFMA compute-bound
        "achieved": round(achieved_tf, 2),
        "peak": FP32_PEAK_TFLOPS,
        "unit": "TFLOP/s",
        "frac": round(achieved_tf / FP32_PEAK_TFLOPS, 4),
        "traffic": traffic,
        "kernel": "k_decim4_fft_mfma_tpl<80> (fused fir+decim+fft+mag)",
        "ms_per_launch": round(ms, 4),
    }


def measure_cpu_baseline(taps1, taps2, decim, fft_len, fixed_sample):
    """oracle chain (the CPU restatement, kind 'port') on all host cores,
    bounded to ~10 s of work."""
    import oracle
    cores = os.cpu_count() or 1
    rng = np.random.default_rng(0x5D5D5D5D)

    n = fixed_sample or (1 << 26)
    x = (rng.uniform(-1, 1, (n, 2)) @ [1, 1j]).astype(np.complex64)
    # repeat passes over the same buffer until >= ~10 s of CPU work
    consumed = 0
    passes = 0
    t0 = time.perf_counter()
    while True:
        _, c = oracle.chain_cf32(taps1, taps2, decim, fft_len, x,
                                 capture=False, nthreads=0)
        consumed += c
        passes += 1
        dt = time.perf_counter() - t0
        if dt >= 10.0 or (fixed_sample and passes >= 1) or passes >= 64:
            break
    msps = consumed / dt / 1e6
    return {
        "value": round(msps, 2),
        "unit": "MSample/s",
        "cores": cores,
        "kind": "port",
        "sample": f"{passes} pass(es) over {n} Complex32 samples through "
                  f"the oracle chain, OpenMP {cores} threads, {dt:.1f}s",
    }


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = max(args.gpus, world)

    if not torch.cuda.is_available():
        raise SystemExit("bench.py needs a HIP device (no CPU fallback)")
    torch.cuda.set_device(local)

    import futuresdr_amd as fa
    if fa.device_count() < 1:
        raise SystemExit("futuresdr_amd sees no HIP device")
    fa.set_device(local)
    lib = fa.lib()
    st = torch.cuda.current_stream()

    td = None
    if world > 1:
        import torch.distributed as td_
        td = td_
        td.init_process_group("nccl")

    beta = fa.kaiser_beta(1e-4)
    taps1 = fa.lowpass_kaiser_n(args.taps, beta, 0.1)
    taps2 = fa.lowpass_kaiser_n(args.taps, beta, 0.11)
    chain = fa.Chain(taps1, taps2, args.decim, args.fft)

    S = args.samples
    y2 = (S + 1 - args.taps + 1 - args.taps) // args.decim
    frames = y2 // args.fft
    prod = frames * args.fft

    d_in = alloc_dev(lib, S * 8)
    fa.fill_uniform_dev(d_in.value, S, seed=1000 + rank,
                        stream=st.cuda_stream)
    mag = torch.empty(prod, dtype=torch.float32, device="cuda")
    # spectrum sink: per-bin EMA over the step's frames, emitting one
    # averaged spectrum per step (MovingAvg(width=fft, decay, history=
    # frames) — the reference sink's output cadence for this batch size)
    avg_sink = fa.MovingAvg(args.fft, 0.1, frames)
    avg_spec = torch.empty(args.fft, dtype=torch.float32, device="cuda")
    gathered = None
    if world > 1:
        payload = prod if args.gather_frames else args.fft
        gathered = torch.empty(world * payload, dtype=torch.float32,
                               device="cuda")

    def step():
        chain.run_dev(d_in.value, S, 0, 0, mag.data_ptr(), prod,
                      stream=st.cuda_stream)
        avg_sink.filter_dev(mag.data_ptr(), prod, avg_spec.data_ptr(),
                            args.fft, stream=st.cuda_stream)
        if td is not None:
            td.all_gather_into_tensor(
                gathered, mag if args.gather_frames else avg_spec)

    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    if td is not None:
        td.barrier()
    torch.cuda.synchronize()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    if td is not None:
        td.barrier()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    if td is not None:  # MAX over ranks
        e = torch.tensor([elapsed], dtype=torch.float64, device="cuda")
        td.all_reduce(e, op=td.ReduceOp.MAX)
        elapsed = float(e.item())

    value = n_gpus * S * args.steps / elapsed / 1e6  # whole-job MSample/s

    if rank == 0:
        roofline = measure_chain_roofline(fa, torch, chain, d_in, S,
                                          taps1, taps2, args.decim,
                                          args.fft, args.traffic_file)
        cpu_baseline = None
        if n_gpus == 1 and not args.skip_cpu_baseline:
            log("measuring CPU baseline (oracle chain, all cores)...")
            cpu_baseline = measure_cpu_baseline(taps1, taps2, args.decim,
                                                args.fft, args.cpu_sample)
        result = {
            "metric": "MSample/s through 127-tap C32 FIR→decim4→"
                      "1k-FFT flowgraph @1/2/4/8 GPU",
            "value": round(value, 2),
            "unit": "MSample/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "f32",
            "data": "synthetic",
            "config": {
                "workload": "fir127_decim4_fft1024",
                "taps1": int(taps1.size),
                "taps2": int(taps2.size),
                "decim": args.decim,
                "fft_len": args.fft,
                "samples_per_step_per_gpu": S,
                "frames_per_step_per_gpu": frames,
                "parallelism": f"{n_gpus} independent channels"
                               + ((" + RCCL all-gather of raw |X|^2 frames"
                                   if args.gather_frames else
                                   " + RCCL all-gather of the averaged "
                                   "spectrum (the MovingAvg sink output)")
                                  if n_gpus > 1 else ""),
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(result), flush=True)

    if td is not None:
        td.destroy_process_group()
    lib.fsdr_dev_free(d_in)


if __name__ == "__main__":
    main()
