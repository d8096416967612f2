"""Window-machinery goldens mined from the reference's own WindowIteratorSpec
(query/src/test/scala/filodb/query/exec/WindowIteratorSpec.scala): literal
sample streams with literal expected per-window outputs through the chunked
iterator — the exact results ChunkedWindowIteratorD produces, filtered of NaN
windows the way the spec filters them."""
import numpy as np
import pytest

from conftest import build_store

SAMPLES_GAUGE = [
    (100000, 1.0), (153000, 2.0), (250000, 3.0), (270000, 4.0), (280000, 5.0),
    (360000, 6.0), (430000, 7.0), (690000, 8.0), (700000, 9.0),
    (710000, float("nan")),   # Prom end-of-time-series marker
]


def _eval(fdb, oracle, samples, start, step, end, window, func,
          kind=None):
    kind = kind if kind is not None else fdb.COL_GAUGE
    st = build_store(fdb, [[[(t, v) for t, v in samples]]], kind=kind)
    q = fdb.make_query(start, step, end, window, func)
    out = oracle.eval_series(st.view(), 0, q, q.num_windows)
    return {start + w * step: out[w] for w in range(q.num_windows)}


@pytest.mark.parametrize("func,expected", [
    # WindowIteratorSpec.scala:180-218 (SumOverTime incl. the NaN marker)
    ("sum", {150000: 1.0, 250000: 5.0, 350000: 12.0, 450000: 13.0,
             750000: 17.0}),
    # :466-502 (AvgOverTime)
    ("avg", {150000: 1.0, 250000: 2.5, 350000: 4.0, 450000: 6.5}),
    # :503-539 (CountOverTime)
    ("count", {150000: 1.0, 250000: 2.0, 350000: 3.0, 450000: 2.0}),
    # :593-629 (MinOverTime)
    ("min", {150000: 1.0, 250000: 2.0, 350000: 3.0, 450000: 6.0}),
    # :630-667 (MaxOverTime)
    ("max", {150000: 1.0, 250000: 3.0, 350000: 5.0, 450000: 7.0}),
])
def test_window_iterator_goldens(fdb, oracle, func, expected):
    ids = {"sum": fdb.FN_SUM_OVER_TIME, "avg": fdb.FN_AVG_OVER_TIME,
           "count": fdb.FN_COUNT_OVER_TIME, "min": fdb.FN_MIN_OVER_TIME,
           "max": fdb.FN_MAX_OVER_TIME}
    end = 1100000 if func == "sum" else 700000
    got = _eval(fdb, oracle, SAMPLES_GAUGE, 50000, 100000, end, 100000,
                ids[func])
    finite = {t: v for t, v in got.items() if not np.isnan(v)}
    assert finite == pytest.approx(expected)


def test_prometheus_rate_golden(fdb, oracle):
    """:219-256 — rate() matching the Prometheus rate function on its sample
    stream, to 1e-10 as the spec asserts."""
    samples = [(1548191486000 + i * 10000, v) for i, v in enumerate(
        [84.0, 152.0, 195.0, 222.0, 245.0, 251.0, 329.0, 374.0, 431.0])]
    expected = {
        1548191496000: 0.34,
        1548191511000: 0.555,
        1548191526000: 0.60375,
        1548191541000: 0.668,
        1548191556000: 1.0357142857142858,
    }
    got = _eval(fdb, oracle, samples, 1548191496000, 15000, 1548191796000,
                300000, fdb.FN_RATE, kind=fdb.COL_COUNTER)
    for t, v in expected.items():
        assert got[t] == pytest.approx(v, abs=1e-10), t


def test_rate_nan_markers_golden(fdb, oracle):
    """:257-285 — NaN end-of-series markers interleaved with counter resets;
    the instant-query rate must equal the spec's 0.5870753512132821."""
    samples = [
        (1614821996000, float("nan")), (1614821996100, 489.0),
        (1614821997000, float("nan")), (1614822566000, 19.0),
        (1614822596000, 26.0), (1614822626000, 26.0), (1614822656000, 26.0),
        (1614822686000, 26.0), (1614822716000, 26.0),
        (1614822717000, float("nan")), (1614822866000, 5.0),
    ]
    got = _eval(fdb, oracle, samples, 1614822880000, 15000, 1614822880000,
                900000, fdb.FN_RATE, kind=fdb.COL_COUNTER)
    assert got[1614822880000] == pytest.approx(0.5870753512132821, abs=1e-12)


def test_last_sample_stale_golden(fdb, oracle):
    """:436-465 — default instant selector: value present at
    time - staleSampleAfterMs (300000, filodb-defaults.conf:604) is returned;
    older samples go stale (NaN)."""
    samples = [(100000, 100.0), (153000, 160.0), (200000, 200.0)]
    got = _eval(fdb, oracle, samples, 100000, 100000, 600000, 300001,
                fdb.FN_LAST)
    finite = {t: v for t, v in got.items() if not np.isnan(v)}
    assert finite == pytest.approx({100000: 100.0, 200000: 200.0,
                                    300000: 200.0, 400000: 200.0,
                                    500000: 200.0})


@pytest.mark.gpu
def test_window_iterator_goldens_on_gpu(fdb, oracle):
    """The same WindowIteratorSpec literal expectations through the GPU
    engine (AGG_NONE [S×W] grid) — reference numbers, not oracle numbers."""
    engine = fdb.Engine(0)
    cases = [
        (fdb.FN_SUM_OVER_TIME, 1100000,
         {150000: 1.0, 250000: 5.0, 350000: 12.0, 450000: 13.0, 750000: 17.0}),
        (fdb.FN_AVG_OVER_TIME, 700000,
         {150000: 1.0, 250000: 2.5, 350000: 4.0, 450000: 6.5}),
        (fdb.FN_COUNT_OVER_TIME, 700000,
         {150000: 1.0, 250000: 2.0, 350000: 3.0, 450000: 2.0}),
        (fdb.FN_MIN_OVER_TIME, 700000,
         {150000: 1.0, 250000: 2.0, 350000: 3.0, 450000: 6.0}),
        (fdb.FN_MAX_OVER_TIME, 700000,
         {150000: 1.0, 250000: 3.0, 350000: 5.0, 450000: 7.0}),
    ]
    st = build_store(fdb, [[[(t, v) for t, v in SAMPLES_GAUGE]]])
    ds = engine.upload(st)
    for func, end, expected in cases:
        q = fdb.make_query(50000, 100000, end, 100000, func)
        nw = q.num_windows
        out = np.empty(nw)
        engine.query(ds, q, out=out)
        finite = {50000 + w * 100000: out[w] for w in range(nw)
                  if not np.isnan(out[w])}
        assert finite == pytest.approx(expected), func

    # rate on the Prometheus fixture
    samples = [(1548191486000 + i * 10000, v) for i, v in enumerate(
        [84.0, 152.0, 195.0, 222.0, 245.0, 251.0, 329.0, 374.0, 431.0])]
    st2 = build_store(fdb, [[[(t, v) for t, v in samples]]],
                      kind=fdb.COL_COUNTER)
    q = fdb.make_query(1548191496000, 15000, 1548191796000, 300000,
                       fdb.FN_RATE)
    out = np.empty(q.num_windows)
    engine.query(engine.upload(st2), q, out=out)
    got = {1548191496000 + w * 15000: out[w] for w in range(q.num_windows)}
    for t, v in {1548191496000: 0.34, 1548191511000: 0.555,
                 1548191526000: 0.60375, 1548191541000: 0.668,
                 1548191556000: 1.0357142857142858}.items():
        assert got[t] == pytest.approx(v, abs=1e-10), t

    # NaN-marker counter instant rate
    samples3 = [
        (1614821996000, float("nan")), (1614821996100, 489.0),
        (1614821997000, float("nan")), (1614822566000, 19.0),
        (1614822596000, 26.0), (1614822626000, 26.0), (1614822656000, 26.0),
        (1614822686000, 26.0), (1614822716000, 26.0),
        (1614822717000, float("nan")), (1614822866000, 5.0),
    ]
    st3 = build_store(fdb, [[[(t, v) for t, v in samples3]]],
                      kind=fdb.COL_COUNTER)
    q = fdb.make_query(1614822880000, 15000, 1614822880000, 900000,
                       fdb.FN_RATE)
    out = np.empty(1)
    engine.query(engine.upload(st3), q, out=out)
    assert out[0] == pytest.approx(0.5870753512132821, abs=1e-12)


# ---------------------------------------------------------------------------
# Cross-series aggregator goldens from AggrOverRangeVectorsSpec.scala
# ---------------------------------------------------------------------------

def _agg_grid(fdb, oracle, series_samples, agg_id, start, step, end):
    """Aligned per-timestamp cross-series aggregation: window=0 makes each
    window exactly one sample, mirroring mapReduce over TransientRows."""
    st = build_store(fdb, [[[(t, v) for t, v in s]] for s in series_samples],
                     groups=[0] * len(series_samples))
    q = fdb.make_query(start, step, end, 0, fdb.FN_SUM_OVER_TIME, agg_id, 1)
    return oracle.query_exec(st.view(), q, st.num_series, q.num_windows)


def test_avg_nan_golden(fdb, oracle):
    """:367-386 — avg of one all-1.0 series and one mostly-NaN series must be
    1.0 everywhere (NaN rows never contribute)."""
    nan = float("nan")
    s1 = [(1541190600 + i * 60, nan) for i in range(5)] + \
         [(1541190900, 1.0), (1541190960, 1.0)]
    s2 = [(1541190600 + i * 60, 1.0) for i in range(7)]
    grid = _agg_grid(fdb, oracle, [s1, s2], fdb.AGG_AVG,
                     1541190600, 60, 1541190960)
    np.testing.assert_allclose(grid, np.ones(7), rtol=1e-12)


def test_stddev_stdvar_nan_golden(fdb, oracle):
    """:387-419 — 11 series (8 all-NaN) at two timestamps; the spec's literal
    stdvar/stddev values."""
    nan = float("nan")
    vals = [(3247.0, 3297.0)] + [(nan, nan)] * 6 + [(5173.0, 5173.0),
            (nan, nan), (11583.0, 11583.0), (nan, nan)]
    series = [[(1, a), (2, b)] for a, b in vals]
    got_var = _agg_grid(fdb, oracle, series, fdb.AGG_STDVAR, 1, 1, 2)
    np.testing.assert_allclose(
        got_var, [12698496.88888889, 12585030.222222222], rtol=1e-12)
    got_dev = _agg_grid(fdb, oracle, series, fdb.AGG_STDDEV, 1, 1, 2)
    np.testing.assert_allclose(
        got_dev, [3563.4950384263, 3547.5386146203], rtol=1e-10)


def test_group_aggregator_golden(fdb, oracle):
    """GroupRowAggregator (aggregator/GroupRowAggregator.scala:12-31): 1 where
    any series contributed a non-NaN row, NaN where all were NaN."""
    nan = float("nan")
    s1 = [(1, 5.0), (2, nan), (3, nan)]
    s2 = [(1, 7.0), (2, 3.0), (3, nan)]
    grid = _agg_grid(fdb, oracle, [s1, s2], fdb.AGG_GROUP, 1, 1, 3)
    assert grid[0] == 1.0 and grid[1] == 1.0 and np.isnan(grid[2])


def test_last_sample_floor_property(fdb, oracle):
    """LastSampleFunctionSpec's validation property: for sweeps of start
    offsets and steps over a sparse random series, every window's value is the
    floor sample (last ts <= wEnd) unless it is older than 5 minutes
    (LastSampleFunctionSpec.scala:138-158; chunked window w = 300000)."""
    rng = np.random.default_rng(99)
    now = 1_600_000_000_000
    # generateRandomRawCounterSeries-like: ~20-25s intervals, 200 samples
    gaps = rng.integers(20_000, 25_000, 200)
    ts = now - 4_000_000 + np.cumsum(gaps).astype(np.int64)
    vs = ts.astype(np.float64)
    # three chunks to exercise the chunk-list rule
    n = len(ts)
    chunks = [[(int(t), float(v)) for t, v in zip(ts[i:i + 70], vs[i:i + 70])]
              for i in range(0, n, 70)]
    st = build_store(fdb, [chunks])
    W = 300_000
    for diff in range(-40_000, 40_001, 12_500):
        for step in (2_000, 7_000, 33_000):
            start = int(ts[60]) + diff
            end = start + 100_000
            q = fdb.make_query(start, step, end, W, fdb.FN_LAST)
            out = oracle.eval_series(st.view(), 0, q, q.num_windows)
            for w in range(q.num_windows):
                cur = start + w * step
                idx = np.searchsorted(ts, cur, side="right") - 1
                if idx < 0 or cur - int(ts[idx]) > W:
                    assert np.isnan(out[w]), (diff, step, w)
                else:
                    assert out[w] == vs[idx], (diff, step, w)
