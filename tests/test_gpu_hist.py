"""GPU histogram pipeline parity vs oracle (config #4 shape)."""
import numpy as np
import pytest

from test_hist import make_hist_store, synth_hist

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def engine(fdb):
    return fdb.Engine(0)


def run_pair(fdb, oracle, engine, series, nb, groups, q):
    st = make_hist_store(fdb, series, nb=nb, groups=groups)
    want_s, want_c, want_q = oracle.query_exec_hist(st.view(), q, nb)
    ds = engine.upload(st)
    nw = q.num_windows
    got_s = np.zeros(q.num_groups * nw * nb)
    got_c = np.zeros(q.num_groups * nw)
    got_q = np.zeros(q.num_groups * nw)
    engine.query_hist(ds, q, nb, out_bucket_sums=got_s, out_counts=got_c,
                      out_quantile=got_q)
    return (got_s, got_c, got_q), (want_s, want_c, want_q)


@pytest.mark.parametrize("case", ["clean", "resets", "many_series"])
def test_hist_rate_sum_quantile(fdb, oracle, engine, case):
    rng = np.random.default_rng(hash(case) % 2**31)
    reset_p = 0.03 if case == "resets" else 0.0
    n_series = 64 if case == "many_series" else 8
    nb = 8 if case != "many_series" else 64
    series = [synth_hist(rng, 120, nb=nb, reset_p=reset_p) for _ in range(n_series)]
    groups = [i % 3 for i in range(n_series)]
    import filodb_amd as f
    start = int(series[0][0][25])
    q = f.make_query(start, 15000, start + 60 * 15000, 300000, f.FN_HIST_RATE,
                     f.AGG_SUM, 3, param=0.99)
    (gs, gc, gq), (ws, wc, wq) = run_pair(fdb, oracle, engine, series, nb, groups, q)
    np.testing.assert_array_equal(gc, wc)
    np.testing.assert_allclose(gs, ws, rtol=1e-9, atol=1e-12)
    np.testing.assert_allclose(gq, wq, rtol=1e-9, atol=1e-12, equal_nan=True)


def test_hist_empty_windows(fdb, oracle, engine):
    rng = np.random.default_rng(4)
    series = [synth_hist(rng, 40, nb=8)]
    import filodb_amd as f
    # windows straddling before/inside/after the data
    start = int(series[0][0][0]) - 100000
    q = f.make_query(start, 60000, start + 30 * 60000, 300000, f.FN_HIST_RATE,
                     f.AGG_SUM, 1, param=0.5)
    (gs, gc, gq), (ws, wc, wq) = run_pair(fdb, oracle, engine, series, 8, [0], q)
    np.testing.assert_array_equal(gc, wc)
    np.testing.assert_allclose(gs, ws, rtol=1e-9, atol=1e-12)
    np.testing.assert_allclose(gq, wq, rtol=1e-9, atol=1e-12, equal_nan=True)
