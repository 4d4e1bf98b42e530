#!/usr/bin/env python3
"""bench.py — BASELINE.json metric: MSample/s of Complex32 input through
the 127-tap FIR -> decim-4 -> 1024-pt FFT chain, on N GPUs of one node.

A step = SOURCE WRITE + one pass of the chain over one HBM-resident batch
of synthetic samples (re/im iid uniform[-1,1), device-generated, seeded).
The source's buffer write (the NullSource analogue,
src/blocks/null_source.rs:53-66) is INSIDE the timed region: every step
re-fills the input batch on-device before the chain consumes it. Source
and chain CAN be pipelined over sub-chunks on two streams
(--pipeline-chunks N, the reference actor model: NullSource running
ahead of the Fir block through the slab's circulating buffers) — but on
this workload both stages share HBM and overlap starves the chain, so
the default is serial (see the step comment). A
separate streaming leg (reported as "streaming" in the JSON, never as
`value`) pushes host chunks through the pinned fsdr_ring (H2D on the copy
stream overlapped with compute) — the PCIe-fed rate. At N>1 each
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
    p.add_argument("--samples", type=int, default=1 << 30,
                   help="chain-input Complex32 samples per step per GPU "
                        "(288 GB HBM3E: 8.6 GB batches amortize launch "
                        "overheads — profiles/throughput_curve_r02.txt)")
    p.add_argument("--fft", type=int, default=1024)
    p.add_argument("--decim", type=int, default=4)
    p.add_argument("--taps", type=int, default=127)
    p.add_argument("--gather-frames", action="store_true",
                   help="all-gather every raw magnitude frame instead of "
                        "the MovingAvg-averaged spectrum")
    p.add_argument("--skip-cpu-baseline", action="store_true")
    p.add_argument("--skip-streaming", action="store_true",
                   help="skip the PCIe-fed ring streaming leg")
    p.add_argument("--skip-config3", action="store_true",
                   help="skip the config-3 (FM resampler chain) leg")
    p.add_argument("--pipeline-chunks", type=int, default=0,
                   help="source/chain pipeline depth per step (0 = auto "
                        "= 1: overlap measured slower — fill and chain "
                        "contend for HBM; see step comment)")
    p.add_argument("--skip-roofline", action="store_true",
                   help="skip the roofline reps (for PMC traffic passes "
                        "whose counters the extra d_null-writing reps "
                        "would pollute)")
    p.add_argument("--streaming-chunk", type=int, default=1 << 24,
                   help="ring chunk size in samples for the streaming leg "
                        "(2^24 = 134 MiB/buffer x4 pinned; measured best "
                        "of 2^22..2^24)")
    p.add_argument("--streaming-chunks", type=int, default=32,
                   help="number of chunks for the streaming leg")
    p.add_argument("--cpu-sample", type=int, default=0,
                   help="fixed CPU-baseline sample size (0 = auto ~10s)")
    p.add_argument("--traffic-file", default=os.path.join(
        REPO, "profiles", "traffic_r02.json"))
    return p.parse_args()


def alloc_dev(lib, bytes_):
    p = ctypes.c_void_p()
    if lib.fsdr_dev_alloc(ctypes.byref(p), bytes_) != 0:
        raise RuntimeError(lib.fsdr_last_error().decode())
    return p


def measure_chain_roofline(fa, torch, chain, d_in, n_samples, taps1,
                           taps2, decim, fft_len, traffic_file,
                           d_mag=0, mag_cap=0):
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
        chain.run_dev(d_in.value, n_samples, 0, 0, d_mag, mag_cap,
                      stream=st.cuda_stream)
    torch.cuda.synchronize()
    reps = 20
    ev0 = torch.cuda.Event(enable_timing=True)
    ev1 = torch.cuda.Event(enable_timing=True)
    ev0.record(st)
    for _ in range(reps):
        # same mag-only shape as the timed step (spectra skipped)
        chain.run_dev(d_in.value, n_samples, 0, 0, d_mag, mag_cap,
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
        "kernel": "k_decim4_fft_mfma_ap_tpl<80> (fused fir+decim+fft+mag; "
                  "all-phase aligned-group staging — halves variant for "
                  "unaligned ring carries)",
        "ms_per_launch": round(ms, 4),
    }


def measure_streaming(fa, lib, chain, taps1, taps2, decim, fft_len,
                      chunk, n_chunks):
    """PCIe-fed leg: host chunks -> pinned fsdr_ring (async H2D on the
    ring's copy stream, overlapped with compute) -> fused chain kernel ->
    MovingAvg mag sink on-device. The host source cost (filling the
    pinned slice, the NullSource analogue) is inside the timed region.
    Reported beside the resident number; never `value`."""
    g_len = taps1.size + taps2.size - 1
    reserved = g_len - 1 + decim * fft_len + decim
    ring = lib.fsdr_ring_create(4, chunk, 8, reserved)
    if not ring:
        raise RuntimeError(lib.fsdr_last_error().decode())
    out_cap = (reserved + chunk) // decim + fft_len
    d_mag = ctypes.c_void_p()
    if lib.fsdr_dev_alloc(ctypes.byref(d_mag), out_cap * 4) != 0:
        raise RuntimeError(lib.fsdr_last_error().decode())
    import concurrent.futures as cf

    import torch
    st = torch.cuda.current_stream()
    hp = ctypes.c_void_p()
    items = ctypes.c_size_t()
    dp = ctypes.c_void_p()
    got = ctypes.c_size_t()
    pool = cf.ThreadPoolExecutor(max_workers=8)

    def fill(base, nbytes):
        # the source's buffer write, parallel over host cores (memset
        # releases the GIL) so the leg measures the PCIe/ring bound, not
        # one core's memset
        nt = 8
        step = (nbytes + nt - 1) // nt
        futs = [pool.submit(ctypes.memset,
                            ctypes.c_void_p(base + i * step), 0,
                            min(step, nbytes - i * step))
                for i in range(nt) if i * step < nbytes]
        for f in futs:
            f.result()

    def push(n):
        total = 0
        for _ in range(n):
            lib.fsdr_ring_writer_acquire(ring, ctypes.byref(hp),
                                         ctypes.byref(items))
            fill(hp.value, chunk * 8)
            lib.fsdr_ring_writer_commit(ring, chunk)
            lib.fsdr_ring_reader_acquire(ring, ctypes.byref(dp),
                                         ctypes.byref(got))
            cons, _prod = chain.run_dev(dp.value, got.value, 0, 0,
                                        d_mag.value, out_cap,
                                        stream=st.cuda_stream)
            lib.fsdr_ring_reader_release_consumed(
                ring, cons, ctypes.c_void_p(st.cuda_stream))
            total += chunk
        return total
    try:
        push(4)  # warmup
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        n = push(n_chunks)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        return {
            "value": round(n / dt / 1e6, 2),
            "unit": "MSample/s",
            "chunk_samples": chunk,
            "chunks": n_chunks,
            "note": "host-fed pinned ring (PCIe-inclusive, host memset "
                    "source in timed region); not comparable to `value` "
                    "(HBM-resident)",
        }
    finally:
        lib.fsdr_dev_free(d_mag)
        lib.fsdr_ring_destroy(ring)
        pool.shutdown(wait=False)


def measure_config3(fa, lib, torch):
    """BASELINE configs[2]: HipFir(127-tap lowpass) -> PolyphaseResampler
    4:1 -> 1024-pt FFT on a resident 2^26-sample batch. Reports MS/s of
    chain input plus the resampler kernel's own roofline block."""
    beta = fa.kaiser_beta(1e-4)
    taps = fa.lowpass_kaiser_n(127, beta, 0.1)
    rtaps = fa.lowpass_kaiser_n(128, beta, 0.2)  # n_taps % interp == 0
    fir = fa.Fir(taps)
    rs = fa.Resampler(1, 4, rtaps)
    fft = fa.Fft(1024)
    S = 1 << 26
    y1 = S + 1 - taps.size
    # resamp_status: prod = ((y1+1-128)*1 - 1)//4 floored to interp mult
    y2 = (y1 + 1 - rtaps.size - 1) // 4
    frames = y2 // 1024
    st = torch.cuda.current_stream()
    d_in = alloc_dev(lib, S * 8)
    d_y1 = alloc_dev(lib, y1 * 8)
    d_y2 = alloc_dev(lib, y2 * 8)
    d_sp = alloc_dev(lib, frames * 1024 * 8)
    try:
        fa.fill_uniform_dev(d_in.value, S, seed=7, stream=st.cuda_stream)

        def step():
            fir.filter_dev(d_in.value, S, d_y1.value, y1,
                           stream=st.cuda_stream)
            rs.filter_dev(d_y1.value, y1, d_y2.value, y2,
                          stream=st.cuda_stream)
            fft.bulk_dev(d_y2.value, d_sp.value, frames,
                         stream=st.cuda_stream)
        for _ in range(3):
            step()
        torch.cuda.synchronize()
        reps = 10
        ev0 = torch.cuda.Event(enable_timing=True)
        ev1 = torch.cuda.Event(enable_timing=True)
        ev0.record(st)
        for _ in range(reps):
            step()
        ev1.record(st)
        torch.cuda.synchronize()
        ms = ev0.elapsed_time(ev1) / reps
        # resampler kernel alone (its roofline)
        ev0.record(st)
        for _ in range(reps):
            rs.filter_dev(d_y1.value, y1, d_y2.value, y2,
                          stream=st.cuda_stream)
        ev1.record(st)
        torch.cuda.synchronize()
        rs_ms = ev0.elapsed_time(ev1) / reps
        rs_prod = (y2 // 1) * 1
        rs_tf = rs_prod * rtaps.size * 4 / (rs_ms * 1e-3) / 1e12
        return {
            "value": round(S * 1e3 / ms / 1e6, 2),
            "unit": "MSample/s",
            "workload": "config3 fm: fir127 -> resamp 4:1 (128 taps) -> "
                        "fft1024, resident 2^26 batch",
            "ms_per_step": round(ms, 4),
            "resampler_roofline": {
                "bound": "mfma",
                "achieved": round(rs_tf, 2),
                "peak": FP32_PEAK_TFLOPS,
                "unit": "TFLOP/s",
                "frac": round(rs_tf / FP32_PEAK_TFLOPS, 4),
                "traffic": None,
                "kernel": "k_decim4_mfma_tpl via the 1:4 resampler route "
                          "(k_resamp_tiled_cf32 for interp>1)",
                "ms_per_launch": round(rs_ms, 4),
            },
        }
    finally:
        for p in (d_in, d_y1, d_y2, d_sp):
            lib.fsdr_dev_free(p)


def measure_cpu_baseline(taps1, taps2, decim, fft_len, fixed_sample):
    """Vectorized oracle chain (kind 'port') on all host cores, bounded to
    ~10 s of work: the fused-tap algorithm (same as the GPU chain) over
    deinterleaved planes, AVX2 FMA via omp simd — a defensible tuned-CPU
    number, not the scalar restatement."""
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
                                 capture=False, nthreads=0, fast=True)
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
                  f"the VECTORIZED oracle chain (fused 253-tap, AVX2 FMA "
                  f"omp simd), OpenMP {cores} threads, {dt:.1f}s",
    }


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = max(args.gpus, world)

    if not torch.cuda.is_available():
        raise SystemExit("bench.py needs a HIP device (no CPU fallback)")
    ndev = torch.cuda.device_count()
    if local >= ndev:  # oversubscribed test runs (2 ranks on a 1-GPU box)
        log(f"rank {rank}: only {ndev} device(s); mapping local {local} "
            f"-> {local % ndev}")
        local = local % ndev
    torch.cuda.set_device(local)

    import futuresdr_amd as fa
    if fa.device_count() < 1:
        raise SystemExit("futuresdr_amd sees no HIP device")
    fa.set_device(local)
    lib = fa.lib()
    st = torch.cuda.current_stream()

    # Initialize RCCL whenever launched under torchrun (even world=1) so
    # the distributed path — nccl init, all_gather_into_tensor, barrier,
    # MAX-reduce — is exercised on hardware by single-GPU runs too.
    td = None
    if world > 1 or "TORCHELASTIC_RUN_ID" in os.environ \
            or os.environ.get("FSDR_FORCE_DIST"):
        import torch.distributed as td_
        td = td_
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29871")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        td.init_process_group("nccl")
        log(f"torch.distributed initialized: rank {rank}/{world} (nccl)")

    beta = fa.kaiser_beta(1e-4)
    taps1 = fa.lowpass_kaiser_n(args.taps, beta, 0.1)
    taps2 = fa.lowpass_kaiser_n(args.taps, beta, 0.11)
    chain = fa.Chain(taps1, taps2, args.decim, args.fft)

    S = args.samples
    # The step is the reference's actor pipeline: the SOURCE fills chunk
    # j+1 (its own stream) while the chain consumes chunk j (compute
    # stream), event-chained exactly like the slab's 4 circulating
    # buffers let NullSource run ahead of the Fir block. Every sample is
    # still written by the source inside the timed region; chunks are
    # independent spans (each loses the <253+4096-sample window tail,
    # ~0.002% of a chunk — the same boundary behavior as unconsumed slab
    # leftovers at stream end).
    # Measured: overlap does NOT pay here — the fill is write-bandwidth
    # bound (~5.7 TB/s) and the chain read-needs ~2.9 TB/s, so running
    # them concurrently over the SHARED HBM starves the chain below its
    # compute rate (pc=4 was ~5% slower than serial). Auto = 1; the
    # pipeline stays available for configs where the consumer is
    # compute-bound with bandwidth slack.
    C = args.pipeline_chunks
    if C <= 0:
        C = 1
    chunk0 = (S // C) & ~4095
    offs, lens = [], []
    off = 0
    for j in range(C):
        n_j = chunk0 if j < C - 1 else S - off
        offs.append(off)
        lens.append(n_j)
        off += n_j
    g1 = args.taps + args.taps - 1

    def chunk_prod(n):
        return max(0, (n + 1 - g1) // args.decim) // args.fft * args.fft

    prods = [chunk_prod(n) for n in lens]
    mag_offs = []
    acc = 0
    for p in prods:
        mag_offs.append(acc)
        acc += p
    prod = acc
    frames = prod // args.fft

    d_in = alloc_dev(lib, S * 8)
    fa.fill_uniform_dev(d_in.value, S, seed=1000 + rank,
                        stream=st.cuda_stream)
    mag = torch.empty(max(prod, 1), dtype=torch.float32, device="cuda")
    # spectrum sink: per-bin EMA over the step's frames, emitting one
    # averaged spectrum per step (MovingAvg(width=fft, decay, history=
    # frames) — the reference sink's output cadence for this batch size)
    avg_sink = fa.MovingAvg(args.fft, 0.1, frames)
    avg_spec = torch.empty(args.fft, dtype=torch.float32, device="cuda")
    gathered = None
    if td is not None:
        payload = prod if args.gather_frames else args.fft
        gathered = torch.empty(world * payload, dtype=torch.float32,
                               device="cuda")

    s_src = torch.cuda.Stream()
    ev_fill = [torch.cuda.Event() for _ in range(C)]
    ev_done = [torch.cuda.Event() for _ in range(C)]
    seed_ctr = [0]

    def step():
        seed_ctr[0] += 1
        seed = seed_ctr[0] * 131 + rank
        for j in range(C):
            # the SOURCE writes chunk j on its own stream (NullSource
            # writes its whole buffer every call — null_source.rs:53-66),
            # after the chain's previous pass over this chunk drained
            s_src.wait_event(ev_done[j])
            fa.fill_uniform_dev(d_in.value + offs[j] * 8, lens[j],
                                seed=seed, offset=offs[j],
                                stream=s_src.cuda_stream)
            ev_fill[j].record(s_src)
            st.wait_event(ev_fill[j])
            chain.run_dev(d_in.value + offs[j] * 8, lens[j], 0, 0,
                          mag.data_ptr() + mag_offs[j] * 4, prods[j],
                          stream=st.cuda_stream)
            ev_done[j].record(st)
            avg_sink.filter_dev(mag.data_ptr() + mag_offs[j] * 4,
                                prods[j], avg_spec.data_ptr(), args.fft,
                                stream=st.cuda_stream)
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
        roofline = None
        if not args.skip_roofline:
            roofline = measure_chain_roofline(fa, torch, chain, d_in, S,
                                              taps1, taps2, args.decim,
                                              args.fft, args.traffic_file,
                                              mag.data_ptr(), prod)
        cpu_baseline = None
        if n_gpus == 1 and not args.skip_cpu_baseline:
            log("measuring CPU baseline (oracle chain, all cores)...")
            cpu_baseline = measure_cpu_baseline(taps1, taps2, args.decim,
                                                args.fft, args.cpu_sample)
        streaming = None
        if n_gpus == 1 and not args.skip_streaming:
            log("measuring PCIe-fed ring streaming leg...")
            streaming = measure_streaming(fa, lib, chain, taps1, taps2,
                                          args.decim, args.fft,
                                          args.streaming_chunk,
                                          args.streaming_chunks)
        config3 = None
        if n_gpus == 1 and not args.skip_config3:
            log("measuring config-3 (FM resampler chain) leg...")
            config3 = measure_config3(fa, lib, torch)
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
                "source_in_timed_region": True,
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
            "streaming": streaming,
            "config3": config3,
        }
        print(json.dumps(result), flush=True)

    if td is not None:
        td.destroy_process_group()
    lib.fsdr_dev_free(d_in)


if __name__ == "__main__":
    main()
