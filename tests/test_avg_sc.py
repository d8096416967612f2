"""Downsample avg over sum+count companion columns.

AvgWithSumAndCountOverTimeFuncD (AggrOverTimeFunctions.scala:820-860):
avg(window) = SumOverTime(sum column) / SumOverTime(count column), each
column summed with the usual NaN-skipping window semantics, divided with
plain IEEE division. The count column rides the chunk's companion slot;
the engine runs the SAME fast scan twice with the directory's value
offsets swapped, then divides elementwise.
"""
import numpy as np
import pytest


def build_sc_store(fdb, rng, n_series=16, n=240, nan_p=0.1):
    st = fdb.ChunkStore()
    data = []
    for _ in range(n_series):
        ts = (100000 + np.arange(n) * 15000
              + rng.integers(-250, 251, n)).astype(np.int64)
        ts = np.maximum.accumulate(ts)
        sums = rng.normal(50, 20, n) * rng.integers(1, 30, n)
        counts = rng.integers(1, 30, n).astype(np.float64)
        if nan_p:
            m = rng.random(n) < nan_p
            sums[m] = np.nan            # a stale row voids the sum sample
        sid = st.add_series(0, fdb.COL_GAUGE)
        st.append_sc(sid, ts, sums, counts)
        data.append((ts, sums, counts))
    st.seal()
    return st, data


def naive_avg(data, q):
    nw = q.num_windows
    out = np.empty(len(data) * nw)
    for s, (ts, sums, counts) in enumerate(data):
        for w in range(nw):
            we = q.start + w * q.step
            ws = we - q.window
            m = (ts >= ws) & (ts <= we)   # startRow = first elem >= wStart
            sv, cv = sums[m], counts[m]
            ssum = np.nansum(sv) if np.sum(~np.isnan(sv)) else np.nan
            csum = np.nansum(cv) if np.sum(~np.isnan(cv)) else np.nan
            if not len(sv):
                ssum = csum = np.nan
            with np.errstate(invalid="ignore", divide="ignore"):
                out[s * nw + w] = ssum / csum
    return out


def test_oracle_avg_sc_vs_naive(fdb, oracle):
    rng = np.random.default_rng(61)
    st, data = build_sc_store(fdb, rng)
    q = fdb.make_query(100000 + 25 * 15000, 15000, 100000 + 235 * 15000,
                       300000, fdb.FN_AVG_OVER_TIME)
    got = oracle.query_exec_avg_sc(st.view(), q, st.num_series, q.num_windows)
    want = naive_avg(data, q)
    np.testing.assert_allclose(got, want, rtol=1e-9, atol=1e-12,
                               equal_nan=True)


def test_append_sc_mixing_rejected(fdb):
    st = fdb.ChunkStore()
    sid = st.add_series(0, fdb.COL_GAUGE)
    ts = np.array([100000], dtype=np.int64)
    st.append(sid, ts, np.array([1.0]))
    with pytest.raises(RuntimeError):   # plain rows then counted rows
        st.append_sc(sid, np.array([115000], dtype=np.int64),
                     np.array([2.0]), np.array([3.0]))
    st2 = fdb.ChunkStore()
    sid2 = st2.add_series(0, fdb.COL_GAUGE)
    st2.append_sc(sid2, ts, np.array([2.0]), np.array([3.0]))
    with pytest.raises(RuntimeError):   # counted rows then a plain append
        st2.append(sid2, np.array([115000], dtype=np.int64), np.array([1.0]))
        st2.seal()


@pytest.mark.gpu
def test_gpu_avg_sc_vs_oracle(fdb, oracle):
    rng = np.random.default_rng(67)
    st, _ = build_sc_store(fdb, rng, n_series=32)
    eng = fdb.Engine(0)
    for step, window in [(15000, 300000), (15000, 70000), (60000, 1800000)]:
        q = fdb.make_query(100000 + 25 * 15000, step, 100000 + 235 * 15000,
                           window, fdb.FN_AVG_OVER_TIME)
        want = oracle.query_exec_avg_sc(st.view(), q, st.num_series,
                                        q.num_windows)
        got = eng.query_avg_sc(eng.upload(st), q)
        np.testing.assert_allclose(got, want, rtol=1e-9, atol=1e-12,
                                   equal_nan=True)


@pytest.mark.gpu
def test_gpu_avg_sc_rejects_plain_dataset(fdb):
    st = fdb.ChunkStore()
    sid = st.add_series(0, fdb.COL_GAUGE)
    ts = np.arange(100000, 100000 + 50 * 15000, 15000, dtype=np.int64)
    st.append(sid, ts, np.ones(50))
    st.seal()
    eng = fdb.Engine(0)
    q = fdb.make_query(400000, 15000, 700000, 300000, fdb.FN_AVG_OVER_TIME)
    with pytest.raises(RuntimeError):
        eng.query_avg_sc(eng.upload(st), q)


def test_oracle_avg_sc_multichunk(fdb, oracle):
    """The oracle composition handles chunked series (both sum passes walk
    chunks); the engine currently restricts avg_sc to single-chunk datasets,
    so this pins the semantics a stream-walk extension must match."""
    rng = np.random.default_rng(83)
    st = fdb.ChunkStore()
    st.set_max_rows(90)                  # 240 rows → 3 chunks per series
    data = []
    for _ in range(8):
        ts = (100000 + np.arange(240) * 15000
              + rng.integers(-250, 251, 240)).astype(np.int64)
        ts = np.maximum.accumulate(ts)
        sums = rng.normal(50, 20, 240) * rng.integers(1, 30, 240)
        counts = rng.integers(1, 30, 240).astype(np.float64)
        sums[rng.random(240) < 0.1] = np.nan
        sid = st.add_series(0, fdb.COL_GAUGE)
        st.append_sc(sid, ts, sums, counts)
        data.append((ts, sums, counts))
    st.seal()
    assert st.num_chunks(0) == 3
    q = fdb.make_query(100000 + 25 * 15000, 15000, 100000 + 235 * 15000,
                       300000, fdb.FN_AVG_OVER_TIME)
    got = oracle.query_exec_avg_sc(st.view(), q, st.num_series, q.num_windows)
    want = naive_avg(data, q)
    np.testing.assert_allclose(got, want, rtol=1e-9, atol=1e-12,
                               equal_nan=True)
