#!/usr/bin/env python3
"""WLAN sync front-end demo (BASELINE configs[4] pieces): the sync-short
autocorrelation metric (rx.rs:73-96) and the SyncLong 64-tap matched
correlator (sync_long.rs:18-50) on a synthesized preamble."""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import futuresdr_amd as fa  # noqa: E402


def main():
    if fa.device_count() < 1:
        raise SystemExit("needs a HIP device (MI355X)")
    r = np.random.default_rng(2)

    def cplx(n):
        return (r.uniform(-1, 1, (n, 2)) @ [1, 1j]).astype(np.complex64)

    stf = cplx(16)
    ltf = cplx(64)
    sig = np.concatenate([0.1 * cplx(300), np.tile(stf, 10),
                          np.tile(ltf, 2), 0.1 * cplx(300)])
    n = sig.size
    # sync-short metric
    delayed = np.concatenate([np.zeros(16, np.complex64), sig[:-16]])
    corr = fa.wlan_moving_sum_host(fa.cmul_conj_host(sig, delayed), 48)[:n]
    power = fa.wlan_moving_sum_host(
        (np.abs(sig) ** 2).astype(np.float32), 64)[:n]
    metric = fa.divide_mag_host(corr, np.maximum(power, 1e-9))
    # longest run above threshold = the STF plateau (the first few window
    # lengths are warmup; SyncShort requires MIN_PLATEAU, sync_short.rs)
    above = metric > 0.7
    runs, cur, best, best_start, start = [], 0, 0, 0, 0
    for i, a in enumerate(above):
        cur = cur + 1 if a else 0
        if cur == 1:
            start = i
        if cur > best:
            best, best_start = cur, start
    print(f"sync-short: longest plateau >0.7 is {best} samples starting "
          f"at {best_start} (preamble at 300..460)")
    # sync-long: matched filter on the LTF
    taps = np.conj(ltf[::-1])
    y, _, p, _ = fa.FirCC(taps).filter(sig, n)
    mags = np.abs(y) ** 2
    top2 = np.sort(np.argsort(mags)[-2:])
    print("sync-long: top-2 correlation peaks at", list(map(int, top2)),
          "(spacing", int(top2[1] - top2[0]), ", expect 64)")


if __name__ == "__main__":
    main()
