"""GPU ↔ oracle parity (runs on a real MI355X via gpurun).

Every case: the HIP engine (filodb_amd, the product path — fails loudly without
a GPU) against the CPU oracle on identical sealed stores. Bar (north_star):
integer-derived results bit-exact, double aggregation ≤1e-9 relative.
"""
import numpy as np
import pytest

from conftest import build_store, synth_counter_series, synth_gauge_series

pytestmark = pytest.mark.gpu

FUNCS = {"rate": 0, "increase": 1, "delta": 2, "sum": 3, "count": 4, "avg": 5,
         "min": 6, "max": 7, "stddev": 8, "stdvar": 9, "changes": 10, "last": 12,
         "present": 13, "timestamp": 14, "zscore": 15}
AGGS = {"sum": 1, "count": 2, "min": 3, "max": 4, "avg": 5,
        "stddev": 8, "stdvar": 9, "group": 10}


@pytest.fixture(scope="module")
def engine(fdb):
    return fdb.Engine(0)


def run_both(fdb, oracle, engine, store, q):
    nw = q.num_windows
    ns = store.num_series
    view = store.view()
    if q.agg_id == 0:
        want = oracle.query_exec(view, q, ns, nw, nthreads=4)
        got = np.empty(ns * nw, dtype=np.float64)
        engine.query(engine.upload(store), q, out=got)
    else:
        want = oracle.query_exec(view, q, ns, nw, nthreads=4)
        got = np.empty(q.num_groups * nw, dtype=np.float64)
        engine.query(engine.upload(store), q, out=got)
    return got, want


def check(got, want, rtol=1e-9):
    np.testing.assert_allclose(got, want, rtol=rtol, atol=1e-12, equal_nan=True)


def counter_store(fdb, n_series=64, n=240, seed=5, reset_p=0.01, nan_p=0.0,
                  chunking=(240,)):
    rng = np.random.default_rng(seed)
    series = []
    for _ in range(n_series):
        ts, vs = synth_counter_series(rng, n, reset_p=reset_p)
        if nan_p:
            vs[rng.random(n) < nan_p] = np.nan
        chunks, i = [], 0
        for c in chunking:
            if i >= n:
                break
            chunks.append([(int(t), float(v)) for t, v in zip(ts[i:i+c], vs[i:i+c])])
            i += c
        if i < n:
            chunks.append([(int(t), float(v)) for t, v in zip(ts[i:], vs[i:])])
        series.append(chunks)
    return build_store(fdb, series, kind=fdb.COL_COUNTER)


def gauge_store(fdb, n_series=64, n=240, seed=9, nan_p=0.1, chunking=(240,)):
    rng = np.random.default_rng(seed)
    series = []
    for _ in range(n_series):
        ts, vs = synth_gauge_series(rng, n, nan_p=nan_p)
        chunks, i = [], 0
        for c in chunking:
            if i >= n:
                break
            chunks.append([(int(t), float(v)) for t, v in zip(ts[i:i+c], vs[i:i+c])])
            i += c
        if i < n:
            chunks.append([(int(t), float(v)) for t, v in zip(ts[i:], vs[i:])])
        series.append(chunks)
    return build_store(fdb, series)  # counter schema like the reference fixture


Q = dict(start=100000 + 40 * 15000, step=15000, window=300000)


def mkq(fdb, func, agg=0, ng=0, **kw):
    p = dict(Q)
    p.update(kw)
    end = p.get("end", p["start"] + 150 * 15000)
    return fdb.make_query(p["start"], p["step"], end, p["window"], func, agg, ng)


@pytest.mark.parametrize("func", ["rate", "increase", "delta"])
@pytest.mark.parametrize("case", ["clean", "resets", "nans", "multichunk"])
def test_counter_funcs(fdb, oracle, engine, func, case):
    kw = dict(reset_p=0.0)
    if case == "resets":
        kw = dict(reset_p=0.05)
    elif case == "nans":
        kw = dict(reset_p=0.02, nan_p=0.05)
    elif case == "multichunk":
        kw = dict(reset_p=0.05, chunking=(80, 80, 80))
    st = counter_store(fdb, seed=hash((func, case)) % 2**31, **kw)
    got, want = run_both(fdb, oracle, engine, st, mkq(fdb, FUNCS[func]))
    check(got, want)


@pytest.mark.parametrize("func", ["sum", "count", "avg", "min", "max",
                                  "stddev", "stdvar", "changes", "last",
                                  "present", "timestamp", "zscore"])
@pytest.mark.parametrize("case", ["raw", "nans", "multichunk", "integral"])
def test_gauge_funcs(fdb, oracle, engine, func, case):
    seed = hash((func, case)) % 2**31
    if case == "raw":
        st = gauge_store(fdb, seed=seed, nan_p=0.0)
    elif case == "nans":
        st = gauge_store(fdb, seed=seed, nan_p=0.15)
    elif case == "multichunk":
        st = gauge_store(fdb, seed=seed, nan_p=0.1, chunking=(100, 100, 40))
    else:  # integral values → DDV-encoded doubles
        st = counter_store(fdb, seed=seed, reset_p=0.0)
    got, want = run_both(fdb, oracle, engine, st, mkq(fdb, FUNCS[func]))
    check(got, want)


@pytest.mark.parametrize("agg", list(AGGS))
def test_group_aggregation(fdb, oracle, engine, agg):
    rng = np.random.default_rng(77)
    n_groups = 7
    series, groups = [], []
    for s in range(100):
        ts, vs = synth_gauge_series(rng, 120, nan_p=0.1)
        series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
        groups.append(s % n_groups)
    st = build_store(fdb, series, groups=groups)
    q = mkq(fdb, FUNCS["sum"], AGGS[agg], n_groups, end=Q["start"] + 60 * 15000)
    got, want = run_both(fdb, oracle, engine, st, q)
    # atomic order differs from the oracle's fold → tolerance, not bit-exact
    check(got, want, rtol=1e-9)


def test_fused_rate_sum_by_group(fdb, oracle, engine):
    """Config #5 shape in miniature: sum by(job)(rate(...[5m]))."""
    rng = np.random.default_rng(123)
    n_groups = 10
    series, groups = [], []
    for s in range(200):
        ts, vs = synth_counter_series(rng, 240, reset_p=0.005)
        series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
        groups.append(s % n_groups)
    st = build_store(fdb, series, groups=groups, kind=fdb.COL_COUNTER)
    q = mkq(fdb, FUNCS["rate"], 1, n_groups)
    got, want = run_both(fdb, oracle, engine, st, q)
    check(got, want)


def test_partial_mode_matches_oracle(fdb, oracle, engine):
    rng = np.random.default_rng(42)
    n_groups = 4
    series, groups = [], []
    for s in range(50):
        ts, vs = synth_gauge_series(rng, 100, nan_p=0.2)
        series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
        groups.append(s % n_groups)
    st = build_store(fdb, series, groups=groups)
    q = mkq(fdb, FUNCS["avg"], AGGS["avg"], n_groups, end=Q["start"] + 40 * 15000)
    nw = q.num_windows
    want_s, want_c = oracle.query_exec(st.view(), q, st.num_series, nw, out_counts=True)
    got_s = np.empty(n_groups * nw)
    got_c = np.empty(n_groups * nw)
    engine.query(engine.upload(st), q, out=got_s, out_counts=got_c)
    check(got_s, want_s)
    np.testing.assert_array_equal(got_c, want_c)


def test_golden_rate_fixture_on_gpu(fdb, oracle, engine):
    """The RateFunctionsSpec reset-at-chunk-boundary fixture through the GPU."""
    from test_oracle_rate import COUNTER_SAMPLES, CHUNK2
    st = build_store(fdb, [[COUNTER_SAMPLES, CHUNK2]])
    end_ts, start_ts = 8213070, 8071950
    q = fdb.make_query(end_ts, 10000, end_ts, end_ts - start_ts, fdb.FN_RATE)
    got = np.empty(1)
    engine.query(engine.upload(st), q, out=got)
    correction = 5201.0
    expected = (909.0 + correction - 4419.0) / (8213000 - 8072000) * 1000
    assert got[0] == pytest.approx(expected, abs=1e-7)


def test_large_parity_checksum(fdb, oracle, engine):
    """Bigger sweep: 4k counter series × 240 rows, full [S×W] grid equality
    (size-independent property for the bench-shaped workload)."""
    st = counter_store(fdb, n_series=4096, seed=2024, reset_p=0.001)
    q = mkq(fdb, FUNCS["rate"])
    got, want = run_both(fdb, oracle, engine, st, q)
    check(got, want)


def test_device_output_pointer(fdb, oracle, engine):
    """out_on_device path: result lands in a torch CUDA tensor (the RCCL
    all-reduce input in multi-GPU runs)."""
    import torch
    rng = np.random.default_rng(3)
    series = []
    for s in range(16):
        ts, vs = synth_counter_series(rng, 100)
        series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
    st = build_store(fdb, series, groups=[0] * 16, kind=fdb.COL_COUNTER)
    q = mkq(fdb, FUNCS["rate"], 1, 1, end=Q["start"] + 20 * 15000)
    nw = q.num_windows
    t_sum = torch.empty(nw, dtype=torch.float64, device="cuda")
    t_cnt = torch.empty(nw, dtype=torch.float64, device="cuda")
    engine.query(engine.upload(st), q, out=t_sum, out_counts=t_cnt, on_device=True)
    engine.synchronize()
    want_s, want_c = oracle.query_exec(st.view(), q, st.num_series, nw, out_counts=True)
    np.testing.assert_allclose(t_sum.cpu().numpy(), want_s, rtol=1e-9, equal_nan=True)
    np.testing.assert_array_equal(t_cnt.cpu().numpy(), want_c)


def test_long_lookback_large_capacity(fdb, oracle, engine):
    """Series spanning several 400-row chunks route to the 1600-row kernel
    tier (long lookbacks); parity must hold there too."""
    st = counter_store(fdb, n_series=64, n=800, seed=31, reset_p=0.01,
                       chunking=(400, 400))
    q = mkq(fdb, FUNCS["rate"], end=Q["start"] + 700 * 15000)
    got, want = run_both(fdb, oracle, engine, st, q)
    check(got, want)
    st2 = gauge_store(fdb, n_series=64, n=800, seed=32, nan_p=0.1,
                      chunking=(300, 300, 200))
    q2 = mkq(fdb, FUNCS["avg"], end=Q["start"] + 700 * 15000)
    got2, want2 = run_both(fdb, oracle, engine, st2, q2)
    check(got2, want2)


def test_stddev_partial_mode_gpu(fdb, oracle, engine):
    """Stddev partials (raw sums + sumsq stacked, counts) via the engine match
    the oracle's — the cross-shard merge contract for AGG_STDDEV/STDVAR."""
    rng = np.random.default_rng(77)
    n_groups = 3
    series, groups = [], []
    for s in range(30):
        ts, vs = synth_gauge_series(rng, 80, nan_p=0.15)
        series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
        groups.append(s % n_groups)
    st = build_store(fdb, series, groups=groups)
    ds = engine.upload(st)
    for agg in ("stddev", "stdvar"):
        q = mkq(fdb, FUNCS["sum"], AGGS[agg], n_groups,
                end=Q["start"] + 40 * 15000)
        nw = q.num_windows
        cells = n_groups * nw
        want_s, want_c = oracle.query_exec(st.view(), q, st.num_series, nw,
                                           out_counts=True)
        got_s = np.empty(2 * cells)
        got_c = np.empty(cells)
        engine.query(ds, q, out=got_s, out_counts=got_c)
        check(got_s, want_s)
        np.testing.assert_array_equal(got_c, want_c)
