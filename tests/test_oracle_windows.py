"""Oracle gauge window functions vs an independent Python/numpy recomputation.

Mirrors the reference's own strategy (AggrOverTimeFunctionsSpec.scala:289-400):
the chunked result must equal a naive recomputation over the same samples —
exactly for counts/min/max, within 1e-9 relative for FP sums (the spec uses
+-1e-7 abs; our bar is tighter per north_star).
"""
import numpy as np
import pytest

from conftest import build_store, synth_gauge_series

REL = 1e-9


def naive_window(ts, vs, chunk_bounds, w_start, w_end, func):
    """Independent restatement of the chunked per-window semantics over decoded
    samples. chunk_bounds: list of (lo, hi) row index ranges per chunk."""
    sum_, count, sqsum, icount = np.nan, np.nan, np.nan, 0
    mn = mx = np.nan
    changes, prev = np.nan, np.nan
    for lo, hi in chunk_bounds:
        cts, cvs = ts[lo:hi], vs[lo:hi]
        m = (cts >= w_start) & (cts <= w_end)
        if not m.any():
            continue
        x = cvs[m]
        nn = x[~np.isnan(x)]
        if func == "sum" or func == "avg":
            cs = np.nan if len(nn) == 0 else nn.sum()
            if not np.isnan(cs) and np.isnan(sum_):
                sum_ = 0.0
            sum_ = sum_ + cs
            icount += len(nn)
        elif func == "count":
            if np.isnan(count):
                count = 0.0
            count += len(nn)
        elif func in ("min", "max"):
            if len(nn):
                mn = np.nanmin([mn, nn.min()]) if not np.isnan(mn) else nn.min()
                mx = np.nanmax([mx, nn.max()]) if not np.isnan(mx) else nn.max()
        elif func in ("stddev", "stdvar", "zscore"):
            cs = np.nan if len(nn) == 0 else nn.sum()
            csq = np.nan if len(nn) == 0 else (nn * nn).sum()
            if not np.isnan(cs) and np.isnan(sum_):
                sum_ = 0.0
            sum_ = sum_ + cs
            if not np.isnan(csq) and np.isnan(sqsum):
                sqsum = 0.0
            sqsum = sqsum + csq
            icount += len(nn)
        elif func == "changes":
            if np.isnan(changes):
                changes = 0.0
            for v in x:
                if not np.isnan(v) and prev != v and not np.isnan(prev):
                    changes += 1
                prev = v
            else:
                pass
            prev = x[-1]
    if func == "sum":
        return sum_
    if func == "count":
        return count
    if func == "avg":
        return sum_ / icount if icount > 0 else (sum_ if np.isnan(sum_) else 0.0)
    if func == "min":
        return mn
    if func == "max":
        return mx
    if func in ("stddev", "stdvar"):
        if icount > 0:
            avg = sum_ / icount
            r = sqsum / icount - avg * avg
            return np.sqrt(r) if func == "stddev" else r
        return sum_ if np.isnan(sum_) else 0.0
    if func == "changes":
        return changes
    if func == "last":
        # LastSampleChunkedFunctionD: raw value of last sample <= wEnd within
        # the window; NaN stale markers propagate (RangeFunction.scala:595-694)
        m = (ts >= w_start) & (ts <= w_end)
        return vs[m][-1] if m.any() else np.nan
    if func == "present":
        # PresentOverTimeChunkedFunctionD (RangeFunction.scala:725-745): 1 for
        # a non-NaN last sample; a NaN stale marker steps back ONE chunk row
        best_t, val = -1, np.nan
        for lo, hi in chunk_bounds:
            cts, cvs = ts[lo:hi], vs[lo:hi]
            if cts[-1] < w_start:
                continue
            idx = np.nonzero(cts <= w_end)[0]
            if len(idx) == 0:
                continue
            e = int(idx[-1])
            t = int(cts[e])
            if t >= w_start and t > best_t:
                if not np.isnan(cvs[e]):
                    best_t, val = t, 1.0
                elif e > 0:
                    best_t, val = t, (np.nan if np.isnan(cvs[e - 1]) else 1.0)
        return val
    if func == "timestamp":
        # TimestampChunkedFunction (RangeFunction.scala:705-723): last ts <=
        # wEnd over the window's chunk list (no window-start bound), seconds
        best_t = -1
        for lo, hi in chunk_bounds:
            cts = ts[lo:hi]
            if cts[-1] < w_start:
                continue
            idx = np.nonzero(cts <= w_end)[0]
            if len(idx) and int(cts[idx[-1]]) > best_t:
                best_t = int(cts[idx[-1]])
        return best_t / 1000.0 if best_t >= 0 else np.nan
    if func == "zscore":
        # ZScoreChunkedFunctionD (AggrOverTimeFunctions.scala:1592-1603) over
        # VarOverTime accumulation; lastSample only from a non-NaN range end
        last = np.nan
        for lo, hi in chunk_bounds:
            cts, cvs = ts[lo:hi], vs[lo:hi]
            if cts[-1] < w_start:
                continue
            m = (cts >= w_start) & (cts <= w_end)
            if m.any():
                e = int(np.nonzero(m)[0][-1])
                if not np.isnan(cvs[e]):
                    last = cvs[e]
        if icount > 0:
            avg = sum_ / icount
            sd = np.sqrt(sqsum / icount - avg * avg)
            with np.errstate(invalid="ignore", divide="ignore"):
                return (last - avg) / sd    # 0/0 -> NaN, x/0 -> inf, as in C
        return sum_ if np.isnan(sum_) else 0.0
    raise ValueError(func)


FUNC_IDS = {"sum": 3, "count": 4, "avg": 5, "min": 6, "max": 7,
            "stddev": 8, "stdvar": 9, "changes": 10, "last": 12,
            "present": 13, "timestamp": 14, "zscore": 15}


@pytest.mark.parametrize("func", list(FUNC_IDS))
@pytest.mark.parametrize("nan_p,nchunks", [(0.0, 1), (0.0, 3), (0.1, 1), (0.1, 3)])
def test_gauge_chunked_vs_naive(fdb, oracle, func, nan_p, nchunks):
    rng = np.random.default_rng(hash((func, nan_p, nchunks)) % 2**31)
    n = 120
    ts, vs = synth_gauge_series(rng, n, step=10000, jitter=400, nan_p=nan_p)
    per = n // nchunks
    chunks, bounds = [], []
    for c in range(nchunks):
        lo = c * per
        hi = n if c == nchunks - 1 else (c + 1) * per
        chunks.append([(int(ts[i]), float(vs[i])) for i in range(lo, hi)])
        bounds.append((lo, hi))
    st = build_store(fdb, [chunks])
    # windowSize=20 samples ≈ 200s, step 3 samples ≈ 30s (spec's sliding recipe)
    start = int(ts[25])
    end = int(ts[-1])
    step, window = 30000, 200000
    q = fdb.make_query(start, step, end, window, FUNC_IDS[func])
    out = oracle.eval_series(st.view(), 0, q, q.num_windows)
    for w in range(q.num_windows):
        w_end = start + w * step
        expected = naive_window(ts, vs, bounds, w_end - window, w_end, func)
        got = out[w]
        if np.isnan(expected):
            assert np.isnan(got), (func, w)
        else:
            assert got == pytest.approx(expected, rel=REL, abs=1e-12), (func, w)


def test_sum_matches_simple_sliding(fdb, oracle):
    # AggrOverTimeFunctionsSpec:289-310 — chunked sum == data.sliding(...).sum
    rng = np.random.default_rng(11)
    n = 100
    ts = (100000 + np.arange(n) * 10000).astype(np.int64)
    vs = rng.random(n) * 100
    st = build_store(fdb, [[[(int(t), float(v)) for t, v in zip(ts, vs)]]], max_rows=50)
    # window = exactly 20 samples: [wEnd-190000, wEnd]
    step, window = 10000 * 5, 190000
    start, end = int(ts[19]), int(ts[-1])
    q = fdb.make_query(start, step, end, window, 3)
    out = oracle.eval_series(st.view(), 0, q, q.num_windows)
    for w in range(q.num_windows):
        i_end = 19 + 5 * w
        expected = vs[i_end - 19:i_end + 1].sum()
        assert out[w] == pytest.approx(expected, rel=REL)


def test_empty_window_emits_nan(fdb, oracle):
    ts = (100000 + np.arange(10) * 1000).astype(np.int64)
    vs = np.arange(10, dtype=np.float64) + 0.5
    st = build_store(fdb, [[[(int(t), float(v)) for t, v in zip(ts, vs)]]])
    q = fdb.make_query(200000, 1000, 205000, 500, 3)  # windows beyond the data
    out = oracle.eval_series(st.view(), 0, q, q.num_windows)
    assert np.isnan(out).all()


def _np_rank_interp(q, vals):
    """calculateRank + sorted interpolation (AggrOverTimeFunctions.scala:400-406)."""
    v = np.sort(vals)
    n = len(v)
    rank = q * (n - 1)
    lower = max(0, int(np.floor(rank)))
    upper = min(n - 1, lower + 1)
    weight = rank - np.floor(rank)
    return v[lower] * (1 - weight) + v[upper] * weight


@pytest.mark.parametrize("nchunks", [1, 3])
def test_quantile_over_time_vs_numpy(fdb, oracle, nchunks):
    """quantile_over_time (QuantileOverTimeChunkedFunctionD,
    AggrOverTimeFunctions.scala:1227-1299): sorted interpolation over the
    window's non-NaN samples; q<0 -> -Inf, q>1 -> +Inf, empty -> NaN."""
    rng = np.random.default_rng(61)
    n = 120
    ts, vs = synth_gauge_series(rng, n, step=10000, jitter=300, nan_p=0.12)
    per = n // nchunks
    chunks, bounds = [], []
    for c in range(nchunks):
        lo, hi = c * per, (n if c == nchunks - 1 else (c + 1) * per)
        chunks.append([(int(ts[i]), float(vs[i])) for i in range(lo, hi)])
    st = build_store(fdb, [chunks])
    # use STORED timestamps (approx-const encoding may shift within ±250)
    tsd = np.concatenate([oracle.decode_longs(st.chunk(0, c)[0])
                          for c in range(nchunks)])
    start, step, window = int(tsd[20]), 30000, 200000
    end = int(tsd[-1]) + step
    for q_param in (-0.5, 0.0, 0.25, 0.5, 0.9, 1.0, 1.5):
        q = fdb.make_query(start, step, end, window, fdb.FN_QUANTILE_OVER_TIME,
                           param=q_param)
        out = oracle.eval_series(st.view(), 0, q, q.num_windows)
        for w in range(q.num_windows):
            w_end = start + w * step
            m = (tsd >= w_end - window) & (tsd <= w_end)
            nn = vs[m][~np.isnan(vs[m])]
            if not m.any():
                assert np.isnan(out[w]), (q_param, w)
            elif q_param < 0:
                assert out[w] == -np.inf, (q_param, w)
            elif q_param > 1:
                assert out[w] == np.inf, (q_param, w)
            elif len(nn) == 0:
                assert np.isnan(out[w]), (q_param, w)
            else:
                assert out[w] == pytest.approx(_np_rank_interp(q_param, nn),
                                               rel=1e-12), (q_param, w)


@pytest.mark.parametrize("nchunks", [1, 3])
def test_mad_over_time_vs_numpy(fdb, oracle, nchunks):
    """median_absolute_deviation_over_time
    (AggrOverTimeFunctions.scala:1248-1330): median of |median - v|."""
    rng = np.random.default_rng(67)
    n = 100
    ts, vs = synth_gauge_series(rng, n, step=10000, jitter=300, nan_p=0.1)
    per = n // nchunks
    chunks = [[(int(ts[i]), float(vs[i]))
               for i in range(c * per, n if c == nchunks - 1 else (c + 1) * per)]
              for c in range(nchunks)]
    st = build_store(fdb, [chunks])
    tsd = np.concatenate([oracle.decode_longs(st.chunk(0, c)[0])
                          for c in range(nchunks)])
    start, step, window = int(tsd[15]), 30000, 250000
    end = int(tsd[-1]) + step
    q = fdb.make_query(start, step, end, window, fdb.FN_MAD_OVER_TIME)
    out = oracle.eval_series(st.view(), 0, q, q.num_windows)
    for w in range(q.num_windows):
        w_end = start + w * step
        m = (tsd >= w_end - window) & (tsd <= w_end)
        nn = vs[m][~np.isnan(vs[m])]
        if len(nn) == 0:
            assert np.isnan(out[w]), w
        else:
            med = _np_rank_interp(0.5, nn)
            expect = _np_rank_interp(0.5, np.abs(med - nn))
            assert out[w] == pytest.approx(expect, rel=1e-12), w


@pytest.mark.parametrize("nchunks", [1, 3])
def test_predict_linear_vs_numpy(fdb, oracle, nchunks):
    """predict_linear (PredictLinearChunkedFunctionD,
    AggrOverTimeFunctions.scala:1496-1554): least-squares extrapolation over
    x=(ts-wEnd)/1000; NaN below 2 samples."""
    rng = np.random.default_rng(71)
    n = 90
    ts, vs = synth_gauge_series(rng, n, step=10000, jitter=300, nan_p=0.1)
    per = n // nchunks
    chunks = [[(int(ts[i]), float(vs[i]))
               for i in range(c * per, n if c == nchunks - 1 else (c + 1) * per)]
              for c in range(nchunks)]
    st = build_store(fdb, [chunks])
    tsd = np.concatenate([oracle.decode_longs(st.chunk(0, c)[0])
                          for c in range(nchunks)])
    start, step, window = int(tsd[12]), 30000, 200000
    end = int(tsd[-1]) + step
    for duration in (60.0, 600.0):
        q = fdb.make_query(start, step, end, window, fdb.FN_PREDICT_LINEAR,
                           param=duration)
        out = oracle.eval_series(st.view(), 0, q, q.num_windows)
        for w in range(q.num_windows):
            w_end = start + w * step
            m = (tsd >= w_end - window) & (tsd <= w_end)
            sel = ~np.isnan(vs[m])
            x = (tsd[m][sel] - w_end) / 1000.0
            y = vs[m][sel]
            if len(y) < 2:
                assert np.isnan(out[w]), (duration, w)
                continue
            cn = len(y)
            cov = (x * y).sum() - x.sum() * y.sum() / cn
            var = (x * x).sum() - x.sum() ** 2 / cn
            slope = cov / var
            intercept = y.sum() / cn - slope * x.sum() / cn
            assert out[w] == pytest.approx(slope * duration + intercept,
                                           rel=1e-9), (duration, w)


def test_rate_over_delta_vs_numpy(fdb, oracle):
    """Delta-temporality rate (RateOverDeltaChunkedFunctionD,
    RateFunctions.scala:424-445): sum of per-sample deltas / window seconds.
    The increase-over-delta arm is plain sum_over_time."""
    rng = np.random.default_rng(83)
    n = 100
    ts, vs = synth_gauge_series(rng, n, step=10000, jitter=300, nan_p=0.1)
    vs = np.abs(vs)                # delta counters carry per-interval counts
    st = build_store(fdb, [[[(int(t), float(v)) for t, v in zip(ts, vs)]]])
    tsd = oracle.decode_longs(st.chunk(0, 0)[0])
    start, step, window = int(tsd[15]), 30000, 200000
    end = int(tsd[-1]) + step
    q = fdb.make_query(start, step, end, window, fdb.FN_RATE_OVER_DELTA)
    out = oracle.eval_series(st.view(), 0, q, q.num_windows)
    for w in range(q.num_windows):
        w_end = start + w * step
        m = (tsd >= w_end - window) & (tsd <= w_end)
        nn = vs[m][~np.isnan(vs[m])]
        if len(nn) == 0:
            assert np.isnan(out[w]), w
        else:
            expect = nn.sum() / window * 1000
            assert out[w] == pytest.approx(expect, rel=1e-12), w
