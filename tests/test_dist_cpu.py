"""CPU (gloo, world_size 2) coverage of the multi-GPU spectrum-join logic.

Config 4 (BASELINE.json) shards 8 independent channels, one per GPU, and
joins 1024-bin magnitude spectra with one all-gather.
The collective wiring used by bench.py at N>1 is exercised here on CPU:
each rank runs the oracle chain on its own seeded channel and all-gathers
batched magnitude frames; rank 0 checks every rank's frames bitwise
against a locally recomputed reference.
"""
import os
import sys

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

FFT = 256
FRAMES = 4


def _channel_mags(rank):
    sys.path.insert(0, REPO)
    import oracle as o
    rng = np.random.default_rng(1000 + rank)
    taps1 = rng.uniform(-1, 1, 127).astype(np.float32)
    taps2 = rng.uniform(-1, 1, 127).astype(np.float32)
    n_in = FRAMES * FFT * 4 + 4 * 127  # enough for FRAMES frames
    inp = (rng.uniform(-1, 1, (n_in, 2)) @ [1, 1j]).astype(np.complex64)
    spectra, _ = o.chain_cf32(taps1, taps2, 4, FFT, inp)
    mags = (np.abs(spectra[:FRAMES * FFT]) ** 2).astype(np.float32)
    return mags


def _worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        mine = torch.from_numpy(_channel_mags(rank))
        gathered = [torch.empty_like(mine) for _ in range(world)]
        dist.all_gather(gathered, mine)
        if rank == 0:
            for r in range(world):
                ref = torch.from_numpy(_channel_mags(r))
                assert torch.equal(gathered[r], ref), f"rank {r} mismatch"
    finally:
        dist.destroy_process_group()


def test_allgather_spectrum_join_gloo():
    port = 29511
    mp.spawn(_worker, args=(2, port), nprocs=2, join=True)


def _avg_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        sys.path.insert(0, REPO)
        import oracle as o
        # the bench's N>1 join payload: the MovingAvg-averaged spectrum
        mags = _channel_mags(rank)
        avg, _, _, _, _ = o.moving_avg(FFT, 0.1, FRAMES, mags, FFT)
        mine = torch.from_numpy(avg)
        gathered = [torch.empty_like(mine) for _ in range(world)]
        dist.all_gather(gathered, mine)
        if rank == 0:
            for rk in range(world):
                m = _channel_mags(rk)
                ref, _, _, _, _ = o.moving_avg(FFT, 0.1, FRAMES, m, FFT)
                assert torch.equal(gathered[rk], torch.from_numpy(ref))
    finally:
        dist.destroy_process_group()


def test_allgather_averaged_spectrum_gloo():
    mp.spawn(_avg_worker, args=(2, 29517), nprocs=2, join=True)
