#!/usr/bin/env python3
"""Synthesize the committed 802.11a preamble+frame IQ fixture
(tests/golden/wlan_frame.npz) for the config-5 end-to-end rx test.

Frame layout (IEEE 802.11a / examples/wlan):
  noise | STS x10 (16-sample short training symbol, 160 samples)
        | GI2 (last 32 of LTS) + LTS + LTS (160 samples)
        | N_SYM OFDM symbols (16-sample CP + 64 samples each)
  ... gap > MIN_GAP ... second frame ... trailing noise.

The long training symbol is synthesized as conj(LONG)/||.||, where LONG
is the reference's correlator tap table (sync_long.rs:188-253, parsed
out of our product source where it is cited) — the matched filter
sum_k x[i+k]*LONG[k] then peaks exactly at LTS alignment. A small CFO
eps is applied to each frame to exercise the coarse (SyncShort) and
fine (SyncLong) frequency-correction paths. Reference data files are
NOT copied; everything here is synthesized.
"""
import os
import re

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
HIP_SRC = os.path.join(HERE, "..", "..", "futuresdr_amd", "csrc",
                       "futuresdr_hip.hip")
OUT = os.path.join(HERE, "wlan_frame.npz")

N_SYM = 8
EPS = 1e-4  # per-sample CFO in radians


def long_taps():
    src = open(HIP_SRC).read()
    blk = src[src.index("WLAN_LONG[64]"):]
    blk = blk[:blk.index("};")]
    vals = re.findall(r"\{([-0-9.]+)f, ([-0-9.]+)f\}", blk)
    assert len(vals) == 64
    return np.array([float(a) + 1j * float(b) for a, b in vals],
                    np.complex64)


def main():
    rng = np.random.default_rng(0x80211A)
    LONG = long_taps()
    lts = (np.conj(LONG) / np.abs(LONG).mean()).astype(np.complex64)

    def cplx(n, scale=1.0):
        return (scale * (rng.uniform(-1, 1, (n, 2)) @ [1, 1j])).astype(
            np.complex64)

    sts = cplx(16)  # 16-sample short training symbol, unit-ish power

    def frame():
        parts = [np.tile(sts, 10), lts[-32:], lts, lts]
        for _ in range(N_SYM):
            sym = cplx(64)
            parts.append(sym[-16:])  # cyclic prefix
            parts.append(sym)
        f = np.concatenate(parts)
        n = np.arange(f.size, dtype=np.float64)
        return (f * np.exp(1j * EPS * n)).astype(np.complex64)

    noise = 0.05
    f1, f2 = frame(), frame()
    # 80 leading ZEROS: the metric warmup is then 0/0 = NaN (no trigger),
    # exactly as in the reference's f32 arithmetic; without them the
    # |abs48|/0 = inf prologue triggers a junk copy in the reference too.
    lead, gap, tail = 700, 1200, 600
    sig = np.concatenate([np.zeros(80, np.complex64), cplx(lead, noise),
                          f1, cplx(gap, noise), f2, cplx(tail, noise)])
    starts = np.array([80 + lead, 80 + lead + f1.size + gap], np.int64)
    np.savez(OUT, iq=sig, preamble_starts=starts, eps=EPS, n_sym=N_SYM,
             long_taps=LONG, lts=lts, frame_len=f1.size)
    print(f"wrote {OUT}: {sig.size} samples, frames at {starts.tolist()}")


if __name__ == "__main__":
    main()
