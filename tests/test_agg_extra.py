"""Cross-series quantile (t-digest) and count_values aggregators.

Quantile parity note: the reference's digest comes from the
third-party com.tdunning:t-digest library, absent from the reference tree, so
tdigest_impl.h restates the PUBLISHED merging algorithm and parity is anchored
on the reference's own test literals (AggrOverRangeVectorsSpec:296-352 —
reproduced exactly below) plus oracle↔engine equality (identical shared code).
Larger digests are approximate by design on both sides.
"""
import numpy as np
import pytest


def _mk_store(fdb, values, ts=(100000, 115000)):
    """values: list of per-series sample tuples aligned with ts."""
    st = fdb.ChunkStore()
    for sv in values:
        sid = st.add_series(0, fdb.COL_GAUGE)
        st.append(sid, np.array(ts, dtype=np.int64),
                  np.array(sv, dtype=np.float64))
    st.seal()
    return st


def _last_query(fdb, agg, ng=1, param=0.0):
    q = fdb.make_query(100000, 15000, 115000, 15000, fdb.FN_LAST, agg, ng,
                       param=param)
    return q


def test_quantile_spec_literals(fdb, oracle):
    """AggrOverRangeVectorsSpec:341-352: quantile(0.5) over
    {NaN,4.6,2.1} -> 3.35 and {5.6,4.4,5.4} -> 5.4."""
    st = _mk_store(fdb, [(np.nan, 5.6), (4.6, 4.4), (2.1, 5.4)])
    q = _last_query(fdb, fdb.AGG_QUANTILE, param=0.5)
    got = oracle.query_exec(st.view(), q, st.num_series, q.num_windows)
    np.testing.assert_allclose(got, [3.35, 5.4], rtol=1e-12)


def test_quantile_spec_literal_070(fdb, oracle):
    """The q=0.70 case (AggrOverRangeVectorsSpec:132-143) against the same
    singleton-centroid interpolation the library applies at small n."""
    st = _mk_store(fdb, [(1.0, 1.0), (2.0, 2.0), (3.0, 3.0), (4.0, 4.0)])
    q = _last_query(fdb, fdb.AGG_QUANTILE, param=0.70)
    got = oracle.query_exec(st.view(), q, st.num_series, q.num_windows)
    # index = 0.7*4 = 2.8: centers at 0.5,1.5,2.5,3.5 -> lerp(3,4,0.3) = 3.3
    np.testing.assert_allclose(got, [3.3, 3.3], rtol=1e-12)


def test_quantile_empty_and_single(fdb, oracle):
    st = _mk_store(fdb, [(np.nan, 7.5)])
    q = _last_query(fdb, fdb.AGG_QUANTILE, param=0.9)
    got = oracle.query_exec(st.view(), q, st.num_series, q.num_windows)
    assert np.isnan(got[0])          # empty digest -> NaN
    assert got[1] == 7.5             # single centroid -> its mean


def test_quantile_monotone_and_bounded(fdb, oracle):
    """Digest quantiles are within [min,max] and monotone in q (sanity over a
    larger digest where results are approximate)."""
    rng = np.random.default_rng(5)
    vals = rng.normal(10, 3, 500)
    st = _mk_store(fdb, [(v, v) for v in vals])
    prev = -np.inf
    for qq in (0.01, 0.25, 0.5, 0.75, 0.99):
        q = _last_query(fdb, fdb.AGG_QUANTILE, param=qq)
        got = oracle.query_exec(st.view(), q, st.num_series, q.num_windows)
        assert vals.min() - 1e-9 <= got[0] <= vals.max() + 1e-9
        assert got[0] >= prev
        prev = got[0]
        # the t-digest tracks the exact quantile closely at these sizes
        assert abs(got[0] - np.quantile(vals, qq)) < 0.5


def test_count_values_oracle(fdb, oracle):
    st = _mk_store(fdb, [(1.0, 2.0), (1.0, 2.0), (3.0, np.nan), (1.0, 2.0)])
    q = _last_query(fdb, fdb.AGG_COUNT_VALUES)
    vals, cnts, n = oracle.count_values(st.view(), q, k_cap=8)
    assert n.tolist() == [2, 1]
    assert vals[0][:2].tolist() == [1.0, 3.0]
    assert cnts[0][:2].tolist() == [3.0, 1.0]
    assert vals[1][0] == 2.0 and cnts[1][0] == 3.0


def test_count_values_limit(fdb, oracle):
    st = _mk_store(fdb, [(float(i), float(i)) for i in range(10)])
    q = _last_query(fdb, fdb.AGG_COUNT_VALUES)
    with pytest.raises(RuntimeError):
        oracle.count_values(st.view(), q, k_cap=4)


@pytest.mark.gpu
class TestGpu:
    @pytest.fixture(scope="class")
    def engine(self, fdb):
        return fdb.Engine(0)

    def test_quantile_gpu_matches_oracle(self, fdb, oracle, engine):
        """Engine t-digest == oracle t-digest (shared implementation, same
        ascending-series insertion order) on a rate-by-group query."""
        from conftest import build_store, synth_counter_series
        rng = np.random.default_rng(2025)
        n_groups = 5
        series, groups = [], []
        for s in range(120):
            ts, vs = synth_counter_series(rng, 240, reset_p=0.01)
            series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
            groups.append(s % n_groups)
        st = build_store(fdb, series, groups=groups, kind=fdb.COL_COUNTER)
        q = fdb.make_query(100000 + 40 * 15000, 15000, 100000 + 150 * 15000,
                           300000, fdb.FN_RATE, fdb.AGG_QUANTILE, n_groups,
                           param=0.75)
        nw = q.num_windows
        want = oracle.query_exec(st.view(), q, st.num_series, nw)
        got = np.empty(n_groups * nw)
        engine.query(engine.upload(st), q, out=got)
        np.testing.assert_allclose(got, want, rtol=1e-12, atol=1e-12,
                                   equal_nan=True)

    def test_quantile_gpu_spec_literals(self, fdb, engine):
        st = _mk_store(fdb, [(np.nan, 5.6), (4.6, 4.4), (2.1, 5.4)])
        q = _last_query(fdb, fdb.AGG_QUANTILE, param=0.5)
        got = np.empty(q.num_windows)
        engine.query(engine.upload(st), q, out=got)
        np.testing.assert_allclose(got, [3.35, 5.4], rtol=1e-12)

    def test_count_values_gpu(self, fdb, oracle, engine):
        from conftest import build_store, synth_gauge_series
        rng = np.random.default_rng(77)
        n_groups = 3
        series, groups = [], []
        for s in range(60):
            ts, vs = synth_gauge_series(rng, 120, nan_p=0.1)
            vs = np.round(vs)        # few distinct values per cell
            series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
            groups.append(s % n_groups)
        st = build_store(fdb, series, groups=groups)
        q = fdb.make_query(100000 + 40 * 15000, 15000, 100000 + 100 * 15000,
                           300000, fdb.FN_LAST, fdb.AGG_COUNT_VALUES, n_groups)
        wv, wc, wn = oracle.count_values(st.view(), q, k_cap=64)
        ds = engine.upload(st)
        gv, gc, gn = engine.count_values(ds, q, k_cap=64)
        np.testing.assert_array_equal(gn, wn)
        for i in range(len(wn)):
            np.testing.assert_array_equal(gv[i][:wn[i]], wv[i][:wn[i]])
            np.testing.assert_array_equal(gc[i][:wn[i]], wc[i][:wn[i]])
