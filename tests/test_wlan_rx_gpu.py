"""WLAN rx front end end-to-end (BASELINE config 5).

The committed fixture (tests/golden/wlan_frame.npz, synthesized — no
reference data files copied) flows through the full rx front end:

  GPU: delay16 / mult-conj / movsum48 / mag^2+movsum64 / divide
       (examples/wlan/src/bin/rx.rs:60-96)
  host: SyncShort state machine (sync_short.rs:92-150)
  GPU:  SyncLong 64-tap LONG correlator (sync_long.rs:18-50)
  host: SyncLong top-2 sync + CP strip (sync_long.rs:136-178)
  GPU:  64-pt Fft block (rx.rs:98)

and is pinned against a pure-Python oracle restatement of the same
reference algorithms (cited below), plus fixture ground truth (2 frames
at known offsets). The state machines consume the same GPU-computed
metric streams in both product and oracle, so threshold decisions are
identical and the comparison isolates the arithmetic."""
import os

import numpy as np
import pytest

pytestmark = []  # per-test marks below

HERE = os.path.dirname(os.path.abspath(__file__))
FIXTURE = os.path.join(HERE, "golden", "wlan_frame.npz")

THRESHOLD = 0.56   # sync_short.rs:4
MIN_GAP = 480      # sync_short.rs:2
MAX_SAMPLES = 540 * 80  # sync_short.rs:3
SEARCH_WINDOW = 320     # sync_long.rs:3


def movsum(x, length):
    """moving_average.rs:65-105: len-1 zeros, then full-window sums."""
    n = x.size
    cs = np.cumsum(np.concatenate([np.zeros(1, x.dtype), x]))
    out = np.zeros(n, x.dtype)
    out[length - 1:] = cs[length:] - cs[:-length][: n - length + 1]
    return out


def metric_chain_numpy(sig):
    """rx.rs:60-96 in f64 numpy (CPU fixture sanity)."""
    delayed = np.concatenate([np.zeros(16, sig.dtype), sig[:-16]])
    mc = sig.astype(np.complex128) * np.conj(delayed.astype(np.complex128))
    abs48 = movsum(mc, 48)
    power = movsum((np.abs(sig.astype(np.complex128)) ** 2), 64)
    with np.errstate(divide="ignore", invalid="ignore"):
        cor = np.abs(abs48) / power
    return delayed, abs48, cor


def sync_short_oracle(sig, abs48, cor):
    """sync_short.rs:92-150 restated (f64 rotations)."""
    n = min(sig.size, abs48.size, cor.size)
    out, tags = [], []
    state, copied, foff, above, pending = 0, 0, 0.0, False, None
    i = 0
    while i < n:
        c = cor[i]
        if state == 0:
            if c > THRESHOLD:
                state = 1
        elif state == 1:
            if c > THRESHOLD:
                foff = -np.angle(abs48[i]) / 16.0
                state, copied, above, pending = 2, 0, False, foff
            else:
                state = 0
        else:
            if c > THRESHOLD:
                if above and copied > MIN_GAP:
                    foff = -np.angle(abs48[i]) / 16.0
                    copied, above, pending = 0, False, foff
                    i += 1
                    continue
                above = True
            else:
                above = False
            if copied == 0 and pending is not None:
                tags.append((len(out), float(pending)))
                pending = None
            out.append(sig[i] * np.exp(1j * foff * copied))
            if copied + 1 == MAX_SAMPLES:
                state = 0
            else:
                copied += 1
        i += 1
    return np.array(out, np.complex64), tags


def sync_long_oracle(x, tags, LONG):
    """sync_long.rs:18-50 (correlator) + :136-178 (Sync/Copy)."""
    xd = x.astype(np.complex128)
    out, frames = [], []
    for t, (T, _f) in enumerate(tags):
        T2 = tags[t + 1][0] if t + 1 < len(tags) else x.size
        if T2 - T < SEARCH_WINDOW + 128:
            continue
        win = xd[T:T + SEARCH_WINDOW + 63]
        cor = np.array([np.dot(win[i:i + 64], LONG)
                        for i in range(SEARCH_WINDOW)])
        mags = np.abs(cor) ** 2
        i0 = int(np.argmax(mags))
        m2 = mags.copy()
        m2[i0] = -1.0
        i1 = int(np.argmax(m2))
        first, second = min(i0, i1), max(i0, i1)
        freq = float(np.angle(cor[first] * np.conj(cor[second])) / 64.0)
        off = first
        idx = np.arange(128)
        out.append(xd[T + off:T + off + 128] * np.exp(1j * idx * freq))
        frames.append((off, freq))
        cur, nc = T + off + 128, 0
        while cur + 80 <= T2:
            k = np.arange(64)
            out.append(xd[cur + 16:cur + 80] *
                       np.exp(1j * (nc * 80 + 128 + 16 + k) * freq))
            cur += 80
            nc += 1
    sy = (np.concatenate(out) if out else np.array([], np.complex128))
    return sy.astype(np.complex64), frames


def rel_l2(a, b):
    d = np.linalg.norm(np.asarray(a, np.complex128) -
                       np.asarray(b, np.complex128))
    n = np.linalg.norm(np.asarray(b, np.complex128))
    return d / max(n, 1e-30)


def test_wlan_fixture_oracle_cpu():
    """CPU-only: the fixture exercises both frames through the oracle
    restatement — pins the fixture shape and the oracle itself."""
    d = np.load(FIXTURE)
    sig = d["iq"]
    delayed, abs48, cor = metric_chain_numpy(sig)
    fs, tags = sync_short_oracle(delayed, abs48, cor)
    # one tag per preamble (the zero prologue keeps the metric warmup
    # NaN, so no junk trigger at stream start)
    assert len(tags) == 2, tags
    sy, frames = sync_long_oracle(fs, tags, d["long_taps"])
    assert len(frames) == 2, (len(frames), tags)
    n_sym = int(d["n_sym"])
    # per frame: 128 preamble + 64 per 80-sample symbol consumed until
    # the next tag (the reference's Copy state keeps converting the gap
    # noise too — FrameEqualizer/decoder discard it downstream)
    assert sy.size >= 2 * (128 + 64 * n_sym), sy.size
    # the two LTS correlation peaks are 64 apart and the chosen offset
    # puts the 128-sample copy on [lts, lts]
    lts = d["lts"].astype(np.complex128)
    two = np.concatenate([lts, lts])
    got0 = sy[:128].astype(np.complex128)
    c = abs(np.vdot(two, got0)) / (np.linalg.norm(two) *
                                   np.linalg.norm(got0))
    assert c > 0.9, c


@pytest.mark.gpu
def test_wlan_rx_end_to_end(gpu):
    """Config 5 end-to-end on GPU + product host state machines, pinned
    against the oracle composition fed the SAME GPU metric streams."""
    d = np.load(FIXTURE)
    sig = d["iq"]
    n = sig.size
    # --- GPU metric chain (rx.rs:60-96) ---
    delayed = np.concatenate([np.zeros(16, np.complex64), sig[:-16]])
    mc = gpu.cmul_conj_host(sig, delayed)
    abs48 = gpu.wlan_moving_sum_host(mc, 48)[:n]
    mag, _, _, _ = gpu.Mag2().filter(sig, n)
    power = gpu.wlan_moving_sum_host(mag.astype(np.float32), 64)[:n]
    cor = gpu.divide_mag_host(abs48, power)
    # --- product host state machines + GPU correlator ---
    rx = gpu.WlanRx()
    fs, tags = rx.sync_short(delayed, abs48, cor)
    sy, frames = rx.sync_long(fs, tags)
    # --- oracle composition on the SAME streams ---
    fs_o, tags_o = sync_short_oracle(delayed.astype(np.complex128),
                                     abs48.astype(np.complex128),
                                     cor.astype(np.float64))
    sy_o, frames_o = sync_long_oracle(fs_o, tags_o, d["long_taps"])
    # integer behavior identical
    assert [t[0] for t in tags] == [t[0] for t in tags_o]
    assert fs.size == fs_o.size
    assert [f[0] for f in frames] == [f[0] for f in frames_o]
    assert sy.size == sy_o.size
    # arithmetic within f32 tolerance
    assert rel_l2(fs, fs_o) < 1e-4
    for (_, f1), (_, f2) in zip(tags, tags_o):
        assert abs(f1 - f2) < 1e-5
    for (_, f1), (_, f2) in zip(frames, frames_o):
        assert abs(f1 - f2) < 1e-4
    assert rel_l2(sy, sy_o) < 1e-3
    # fixture ground truth: two frames, full symbol payload each
    n_sym = int(d["n_sym"])
    assert len(frames) == 2
    assert sy.size >= 2 * (128 + 64 * n_sym)
    # --- GPU 64-pt FFT of the symbol stream (rx.rs:98) ---
    fft = gpu.Fft(64)
    outs = []
    off = 0
    while off < sy.size:
        y, c, p, _ = fft.filter(sy[off:], sy.size - off)
        assert p > 0
        outs.append(y)
        off += c
    spec = np.concatenate(outs)
    ref = np.fft.fft(
        sy_o.astype(np.complex128).reshape(-1, 64), axis=1).ravel()
    assert rel_l2(spec, ref) < 1e-3
