"""Histogram max/min companion columns (otel schema):
SumAndMaxOverTimeFuncHD and CumulativeHistRateAndMinMaxFunction
(AggrOverTimeFunctions.scala:612-813), HistMaxMinSumAggregator merges.
"""
import numpy as np
import pytest

from test_hist import synth_hist


def make_mm_store(fdb, series, nb=8, groups=None):
    """series: list of (ts, cum, maxs, mins) tuples (one chunk each)."""
    st = fdb.ChunkStore()
    for i, (ts, bv, mx, mn) in enumerate(series):
        gid = groups[i] if groups else 0
        sid = st.add_series(gid, fdb.COL_HIST)
        st.append_hist_mm(sid, ts, bv, mx, mn, first=2.0, mult=2.0)
    st.seal()
    return st


def synth_mm(rng, n, nb=8, reset_p=0.0, start_ts=100000):
    ts, cum = synth_hist(rng, n, nb=nb, reset_p=reset_p, start_ts=start_ts)
    maxs = rng.random(n) * 50
    mins = rng.random(n) * 5
    maxs[rng.random(n) < 0.1] = np.nan
    mins[rng.random(n) < 0.1] = np.nan
    return ts, cum, maxs, mins


def naive_minmax(series, groups, q, ng):
    """NaN-ignoring per-window max/min over the companion columns, merged
    across series (single-chunk series: plain row-range rules)."""
    nw = q.num_windows
    gmax = np.full((ng, nw), np.nan)
    gmin = np.full((ng, nw), np.nan)
    for (ts, _, mx, mn), g in zip(series, groups):
        ts = np.asarray(ts)
        for w in range(nw):
            wEnd = q.start + w * q.step
            wStart = wEnd - q.window
            sel = (ts >= wStart) & (ts <= wEnd)
            if sel.any():
                m = mx[sel]
                if not np.all(np.isnan(m)):
                    v = np.nanmax(m)
                    if np.isnan(gmax[g, w]) or v > gmax[g, w]:
                        gmax[g, w] = v
                m = mn[sel]
                if not np.all(np.isnan(m)):
                    v = np.nanmin(m)
                    if np.isnan(gmin[g, w]) or v < gmin[g, w]:
                        gmin[g, w] = v
    return gmax.ravel(), gmin.ravel()


@pytest.fixture()
def mm_case(fdb):
    rng = np.random.default_rng(66)
    nb = 8
    series = [synth_mm(rng, 120, nb=nb, reset_p=0.02) for _ in range(10)]
    groups = [i % 3 for i in range(10)]
    st = make_mm_store(fdb, series, nb=nb, groups=groups)
    start = int(series[0][0][20])
    q = fdb.make_query(start, 15000, start + 80 * 15000, 300000,
                       fdb.FN_HIST_RATE, fdb.AGG_SUM, 3, param=0.9)
    return st, q, series, groups, nb


def test_oracle_rate_minmax_vs_naive(fdb, oracle, mm_case):
    st, q, series, groups, nb = mm_case
    sums, cnts, mx, mn, quant = oracle.query_exec_hist_mm(st.view(), q, nb)
    wmax, wmin = naive_minmax(series, groups, q, 3)
    np.testing.assert_allclose(mx, wmax, rtol=1e-12, equal_nan=True)
    np.testing.assert_allclose(mn, wmin, rtol=1e-12, equal_nan=True)
    # the rate/quantile legs must equal the plain hist pipeline
    s2, c2, q2 = oracle.query_exec_hist(st.view(), q, nb)
    np.testing.assert_allclose(sums, s2, rtol=1e-12)
    np.testing.assert_array_equal(cnts, c2)


def test_oracle_hist_sum_vs_naive(fdb, oracle, mm_case):
    st, q, series, groups, nb = mm_case
    q.func_id = fdb.FN_SUM_OVER_TIME
    sums, cnts, mx, mn, _ = oracle.query_exec_hist_mm(st.view(), q, nb)
    nw = q.num_windows
    want = np.zeros((3, nw, nb))
    for (ts, cum, _, _), g in zip(series, groups):
        ts = np.asarray(ts)
        for w in range(nw):
            wEnd = q.start + w * q.step
            sel = (ts >= wEnd - q.window) & (ts <= wEnd)
            if sel.any():
                want[g, w] += cum[sel].sum(axis=0)
    np.testing.assert_allclose(sums, want.ravel(), rtol=1e-9)


@pytest.mark.gpu
class TestGpuMM:
    @pytest.fixture(scope="class")
    def engine(self, fdb):
        return fdb.Engine(0)

    @pytest.mark.parametrize("func", ["rate", "sum"])
    def test_gpu_matches_oracle(self, fdb, oracle, engine, func):
        rng = np.random.default_rng(hash(func) % 2**31)
        nb = 16
        # multi-chunk series too: two appended chunk groups
        series = [synth_mm(rng, 150, nb=nb, reset_p=0.02) for _ in range(12)]
        groups = [i % 4 for i in range(12)]
        st = fdb.ChunkStore()
        st.set_max_rows(80)          # forces 2 chunks per series
        for i, (ts, bv, mx, mn) in enumerate(series):
            sid = st.add_series(groups[i], fdb.COL_HIST)
            st.append_hist_mm(sid, ts, bv, mx, mn)
        st.seal()
        start = int(series[0][0][10])
        fid = fdb.FN_HIST_RATE if func == "rate" else fdb.FN_SUM_OVER_TIME
        q = fdb.make_query(start, 15000, start + 100 * 15000, 300000,
                           fid, fdb.AGG_SUM, 4, param=0.9)
        ws, wc, wmx, wmn, wq = oracle.query_exec_hist_mm(st.view(), q, nb)
        nw = q.num_windows
        gs = np.zeros(4 * nw * nb)
        gc = np.zeros(4 * nw)
        gmx = np.zeros(4 * nw)
        gmn = np.zeros(4 * nw)
        gq = np.zeros(4 * nw)
        engine.query_hist_mm(engine.upload(st), q, nb, out_bucket_sums=gs,
                             out_counts=gc, out_max=gmx, out_min=gmn,
                             out_quantile=gq)
        np.testing.assert_array_equal(gc, wc)
        np.testing.assert_allclose(gs, ws, rtol=1e-9, atol=1e-12)
        np.testing.assert_allclose(gmx, wmx, rtol=1e-12, equal_nan=True)
        np.testing.assert_allclose(gmn, wmn, rtol=1e-12, equal_nan=True)
        np.testing.assert_allclose(gq, wq, rtol=1e-9, atol=1e-12,
                                   equal_nan=True)
