"""Property-based round-trip and parity fuzzing (hypothesis).

Mirrors the reference's ScalaCheck layer (EncodingPropertiesTest.scala): any
valid sample stream must encode → decode bit-exactly, and the chunked window
engine must match the naive recomputation for arbitrary windows.
"""
import numpy as np
from hypothesis import given, settings, strategies as st

from conftest import build_store

import filodb_amd as fdb_mod

# FDB_FUZZ_SCALE multiplies example counts for one-off deep runs (default 1)
_SCALE = int(__import__("os").environ.get("FDB_FUZZ_SCALE", "1"))


def _oracle():
    import sys
    import os
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "oracle"))
    import pyclient
    return pyclient


ts_strategy = st.lists(st.integers(min_value=0, max_value=10_000), min_size=3,
                       max_size=120)
val_strategy = st.lists(
    st.one_of(st.floats(min_value=-1e12, max_value=1e12, allow_nan=False),
              st.integers(min_value=-10**12, max_value=10**12).map(float),
              st.just(float("nan"))),
    min_size=3, max_size=120)


@settings(max_examples=60 * _SCALE, deadline=None)
@given(deltas=ts_strategy, vals=val_strategy)
def test_encode_decode_roundtrip(deltas, vals):
    """Arbitrary nondecreasing timestamps + arbitrary doubles (NaNs included)
    survive the encoder bit-exactly — whatever encoding gets chosen, EXCEPT
    the approx-const timestamp loss the reference accepts by design
    (DeltaDeltaVector.scala:46-47): when that triggers, decoded timestamps lie
    on the slope line within ±250."""
    oracle = _oracle()
    n = min(len(deltas), len(vals))
    ts = np.cumsum(np.array(deltas[:n], dtype=np.int64)) + 100_000
    vs = np.array(vals[:n], dtype=np.float64)
    st_ = build_store(fdb_mod, [[[(int(t), float(v)) for t, v in zip(ts, vs)]]],
                      kind=fdb_mod.COL_GAUGE, max_rows=400)
    tsb, vab, nrows, t0, t1 = st_.chunk(0, 0)
    assert nrows == n and t0 == ts[0] and t1 == ts[-1]
    dec_ts = oracle.decode_longs(tsb)
    info = oracle.vec_info(tsb)
    if info["wf"] == 0x0608 and not np.array_equal(dec_ts, ts):
        assert np.abs(dec_ts - ts).max() <= 250   # approx-const band
        assert dec_ts[0] == ts[0]
    else:
        np.testing.assert_array_equal(dec_ts, ts)
    dec_vs = oracle.decode_doubles(vab)
    vinfo = oracle.vec_info(vab)
    if vinfo["wf"] != 0x0506:
        # integral doubles ride the DeltaDelta-long path, where the reference
        # itself canonicalizes -0.0 to +0.0 (LongDoubleWrapper.toLong,
        # DoubleVector.scala:517-538)
        vs = np.where(vs == 0.0, 0.0, vs)
    np.testing.assert_array_equal(
        dec_vs.view(np.uint64), vs.view(np.uint64))  # bit-exact incl. NaNs


@settings(max_examples=50 * _SCALE, deadline=None)
@given(deltas=ts_strategy, vals=val_strategy,
       nchunks=st.integers(min_value=1, max_value=4),
       window=st.integers(min_value=1, max_value=20_000),
       step=st.integers(min_value=1, max_value=5_000),
       func=st.sampled_from(["sum", "count", "avg", "min", "max",
                             "stddev", "stdvar", "changes", "last",
                             "present", "timestamp", "zscore"]))
def test_gauge_window_parity_fuzz(deltas, vals, nchunks, window, step, func):
    """Gauge family under arbitrary samples/NaNs/chunking/windows: the oracle's
    chunked evaluation equals the naive per-window model (the same one
    test_oracle_windows pins against AggrOverTimeFunctionsSpec.scala:289-400),
    including the NaN-poison sum, started-count and chunk-boundary semantics."""
    from test_oracle_windows import naive_window, FUNC_IDS
    oracle = _oracle()
    n = min(len(deltas), len(vals))
    if n < 3:
        return
    ts = np.cumsum(np.array(deltas[:n], dtype=np.int64) + 1) + 100_000
    vs = np.array(vals[:n], dtype=np.float64)
    per = max(1, n // nchunks)
    chunks, bounds = [], []
    for c in range(nchunks):
        lo = c * per
        hi = n if c == nchunks - 1 else min(n, (c + 1) * per)
        if lo >= hi:
            break
        chunks.append([(int(ts[i]), float(vs[i])) for i in range(lo, hi)])
        bounds.append((lo, hi))
    st_ = build_store(fdb_mod, [chunks], kind=fdb_mod.COL_GAUGE, max_rows=400)
    # use STORED timestamps (approx-const encoding may shift them within ±250)
    dec = np.concatenate([oracle.decode_longs(st_.chunk(0, c)[0])
                          for c in range(len(chunks))])
    start = int(dec[0])
    end = int(dec[-1]) + step
    q = fdb_mod.make_query(start, step, end, window, FUNC_IDS[func])
    out = oracle.eval_series(st_.view(), 0, q, q.num_windows)
    for w in range(q.num_windows):
        w_end = start + w * step
        expected = naive_window(dec, vs, bounds, w_end - window, w_end, func)
        got = out[w]
        if np.isnan(expected):
            assert np.isnan(got), (func, w)
        else:
            assert got == expected or abs(got - expected) <= 1e-9 * abs(expected), \
                (func, w, got, expected)


@settings(max_examples=40 * _SCALE, deadline=None)
@given(deltas=ts_strategy,
       incs=st.lists(st.integers(min_value=0, max_value=1000), min_size=3,
                     max_size=120),
       resets=st.lists(st.booleans(), min_size=3, max_size=120),
       window=st.integers(min_value=1, max_value=20_000),
       step=st.integers(min_value=1, max_value=5_000))
def test_rate_window_parity_fuzz(deltas, incs, resets, window, step):
    """Counter series with arbitrary resets: the oracle's windowed rate equals
    a naive recomputation (corrected first/last + extrapolatedRate) for every
    window — the oracle is the GPU's reference, the naive model is its check."""
    oracle = _oracle()
    n = min(len(deltas), len(incs), len(resets))
    if n < 3:
        return
    ts = np.cumsum(np.array(deltas[:n], dtype=np.int64) + 1) + 100_000
    vals = np.zeros(n)
    cur = 0.0
    for i in range(n):
        if resets[i]:
            cur = 0.0
        cur += incs[i]
        vals[i] = cur
    st_ = build_store(fdb_mod, [[[(int(t), float(v)) for t, v in zip(ts, vals)]]],
                      kind=fdb_mod.COL_COUNTER)
    tsb, _, _, _, _ = st_.chunk(0, 0)
    enc_ts = oracle.decode_longs(tsb)
    start = int(ts[0])
    end = int(ts[-1]) + step
    q = fdb_mod.make_query(start, step, end, window, fdb_mod.FN_RATE)
    out = oracle.eval_series(st_.view(), 0, q, q.num_windows)

    # naive corrected series (CorrectingDoubleVectorReader semantics)
    corrected = vals.copy()
    corr = 0.0
    last = -np.inf
    for i in range(n):
        x = vals[i]
        if x < last:
            corr += last
        corrected[i] = x + corr
        last = x
    for w in range(q.num_windows):
        w_end = start + w * step
        w_start = w_end - window
        m = (enc_ts >= w_start) & (enc_ts <= w_end)
        if m.sum() < 2 or enc_ts[m][-1] <= enc_ts[m][0]:
            assert np.isnan(out[w]), w
            continue
        idx = np.nonzero(m)[0]
        expect = oracle.extrapolated_rate(
            int(w_start), int(w_end), int(m.sum()),
            int(enc_ts[idx[0]]), float(corrected[idx[0]]),
            int(enc_ts[idx[-1]]), float(corrected[idx[-1]]), True, True)
        assert out[w] == expect or abs(out[w] - expect) <= 1e-9 * abs(expect), w


@settings(max_examples=40 * _SCALE, deadline=None)
@given(rows=st.lists(
           st.lists(st.integers(min_value=0, max_value=5000), min_size=4,
                    max_size=4),
           min_size=2, max_size=80),
       reset_at=st.lists(st.integers(min_value=1, max_value=79), max_size=4))
def test_hist_encode_decode_fuzz(rows, reset_at):
    """Arbitrary cumulative-LE histogram streams with arbitrary resets must
    survive the sect-delta encoder bit-exactly, with TypeDrop corrections
    equal to the raw value before each drop (SectDeltaHistogramReader
    semantics; drop rule of DeltaSectDiffPackSink, HistogramVector.scala:
    491-545)."""
    oracle = _oracle()
    from test_hist import make_hist_store
    n = len(rows)
    cum = np.cumsum(np.array(rows, dtype=np.uint64), axis=1)  # cumulative-LE
    running = cum.copy()
    for i in range(1, n):
        if i not in reset_at:
            running[i] = running[i - 1] + cum[i]   # keep counters increasing
    ts = (100000 + np.arange(n) * 15000).astype(np.int64)
    st_ = make_hist_store(fdb_mod, [(ts, running)], nb=4)
    _, vab, _, _, _ = st_.chunk(0, 0)
    dec = oracle.hist_decode(vab)
    np.testing.assert_array_equal(dec, running.astype(np.int64))
    # corrections: walk the decoded values with the reference's rule — a new
    # section starts TypeDrop iff some bucket-consecutive delta decreased
    corr = oracle.hist_corrections(vab)
    assert corr.shape == dec.shape
    # invariant: corrected values are nondecreasing over time per bucket
    corrected = dec + corr
    assert (np.diff(corrected, axis=0) >= 0).all()
