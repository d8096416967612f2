"""Cross-series fastReduce parity: oracle group aggregation vs numpy.

Semantics: RangeVectorAggregator.fastReduce (AggrOverRangeVectors.scala:320-377)
with the RowAggregator implementations (aggregator/SumRowAggregator.scala:12-34,
CountRowAggregator.scala:36-42, Min/MaxRowAggregator, AvgRowAggregator.scala:8-41).
"""
import numpy as np
import pytest

from conftest import build_store, synth_gauge_series

AGGS = {"sum": 1, "count": 2, "min": 3, "max": 4, "avg": 5,
        "stddev": 8, "stdvar": 9, "group": 10}


def make_multi(fdb, rng, n_series=40, n_groups=5, n=60, nan_p=0.15):
    series, groups = [], []
    all_ts, all_vs = [], []
    for s in range(n_series):
        ts, vs = synth_gauge_series(rng, n, step=10000, jitter=400, nan_p=nan_p)
        series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
        groups.append(s % n_groups)
        all_ts.append(ts)
        all_vs.append(vs)
    st = build_store(fdb, series, groups=groups)
    return st, groups, all_ts, all_vs


@pytest.mark.parametrize("agg", list(AGGS))
def test_group_reduce_vs_numpy(fdb, oracle, agg):
    rng = np.random.default_rng(17)
    n_groups = 5
    st, groups, all_ts, all_vs = make_multi(fdb, rng, n_groups=n_groups)
    start = 100000 + 20 * 10000
    q = fdb.make_query(start, 30000, start + 300000, 150000, fdb.FN_SUM_OVER_TIME,
                       AGGS[agg], n_groups)
    nw = q.num_windows
    view = st.view()
    grid = oracle.query_exec(view, q, st.num_series, nw)

    # per-series results first (already covered by window tests), then reduce in numpy
    qs = fdb.make_query(q.start, q.step, q.end, q.window, q.func_id)
    per = np.stack([oracle.eval_series(view, s, qs, nw) for s in range(st.num_series)])
    for g in range(n_groups):
        rows = per[np.array(groups) == g]
        for w in range(nw):
            col = rows[:, w]
            nn = col[~np.isnan(col)]
            if agg == "sum":
                exp = np.nan if len(nn) == 0 else nn.sum()
            elif agg == "count":
                exp = np.nan if len(nn) == 0 else float(len(nn))
            elif agg == "min":
                exp = np.nan if len(nn) == 0 else nn.min()
            elif agg == "max":
                exp = np.nan if len(nn) == 0 else nn.max()
            elif agg == "avg":
                exp = np.nan if len(nn) == 0 else nn.mean()
            elif agg == "stddev":
                exp = np.nan if len(nn) == 0 else np.sqrt(
                    (nn * nn).mean() - nn.mean() ** 2)
            elif agg == "group":
                exp = np.nan if len(nn) == 0 else 1.0
            else:  # stdvar
                exp = np.nan if len(nn) == 0 else (nn * nn).mean() - nn.mean() ** 2
            got = grid[g * nw + w]
            if np.isnan(exp):
                assert np.isnan(got)
            else:
                assert got == pytest.approx(exp, rel=1e-9), (g, w)


def test_partial_mode_merge_equals_presented(fdb, oracle):
    """Partial (sum,count) grids merged across two halves must equal the
    presented single-pass result — the ReduceAggregateExec merge contract
    (AggrOverRangeVectors.scala:18-60) used by the multi-GPU all-reduce."""
    rng = np.random.default_rng(23)
    n_groups = 4
    st, groups, _, _ = make_multi(fdb, rng, n_series=30, n_groups=n_groups)
    start = 100000 + 20 * 10000
    for agg in ("sum", "avg", "count"):
        q = fdb.make_query(start, 30000, start + 200000, 100000,
                           fdb.FN_AVG_OVER_TIME, AGGS[agg], n_groups)
        nw = q.num_windows
        presented = oracle.query_exec(st.view(), q, st.num_series, nw)
        sums, counts = oracle.query_exec(st.view(), q, st.num_series, nw,
                                         out_counts=True)
        merged = np.where(counts > 0,
                          (sums / counts) if agg == "avg" else sums,
                          np.nan)
        np.testing.assert_allclose(merged, presented, rtol=1e-12, equal_nan=True)


def test_stddev_partial_two_shard_merge(fdb, oracle):
    """Stddev/stdvar partials = (raw sums, raw sumsq, counts); merging two
    shards by addition and presenting sqrt(sq/n - mean²) must equal the
    single-shard presented result (StddevRowAggregator.scala:36-52)."""
    rng = np.random.default_rng(41)
    n_groups = 3
    series, groups = [], []
    for s in range(24):
        ts, vs = synth_gauge_series(rng, 50, step=10000, jitter=300, nan_p=0.1)
        series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
        groups.append(s % n_groups)
    start = 100000 + 15 * 10000
    for agg in ("stddev", "stdvar"):
        q = fdb.make_query(start, 30000, start + 200000, 120000,
                           fdb.FN_SUM_OVER_TIME, AGGS[agg], n_groups)
        nw = q.num_windows
        cells = n_groups * nw
        # whole dataset, presented single-pass
        st_all = build_store(fdb, series, groups=groups)
        presented = oracle.query_exec(st_all.view(), q, st_all.num_series, nw)
        # two shards: first half / second half of the series
        tot_s = np.zeros(2 * cells)
        tot_c = np.zeros(cells)
        for half in (slice(0, 12), slice(12, 24)):
            st = build_store(fdb, series[half], groups=groups[half])
            sums, counts = oracle.query_exec(st.view(), q, st.num_series, nw,
                                             out_counts=True)
            assert sums.shape == (2 * cells,)
            tot_s += sums
            tot_c += counts
        merged = np.full(cells, np.nan)
        m = tot_c > 0
        mean = tot_s[:cells][m] / tot_c[m]
        var = tot_s[cells:][m] / tot_c[m] - mean * mean
        merged[m] = np.sqrt(var) if agg == "stddev" else var
        np.testing.assert_allclose(merged, presented, rtol=1e-9, atol=1e-12,
                                   equal_nan=True)
