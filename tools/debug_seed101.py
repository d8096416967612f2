"""Reproduce test_randomized_matrix_vs_oracle[101] trial-by-trial with diffs."""
import sys, numpy as np
sys.path.insert(0, "/root/repo/tests")
sys.path.insert(0, "/root/repo")
import filodb_amd as fdb
from oracle import pyclient as oracle
from conftest import build_store, synth_counter_series, synth_gauge_series

engine = fdb.Engine(0)
rng = np.random.default_rng(101)
funcs = [0,1,2,3,4,5,6,7,8,9,10,12,13,14,15,16,17,18,19]
for trial in range(6):
    fid = int(rng.choice(funcs))
    n = int(rng.integers(5, 500))
    nchunks = min(int(rng.integers(1, 7)), n - 1)
    cuts = sorted(rng.choice(np.arange(1, n), size=max(0, nchunks-1), replace=False).tolist()) if nchunks > 1 else []
    series = []
    for _ in range(int(rng.integers(2, 10))):
        if fid <= 2 or fid == 19:
            ts, vs = synth_counter_series(rng, n, reset_p=0.02)
            if rng.random() < 0.3:
                vs[rng.random(n) < 0.05] = np.nan
        else:
            ts, vs = synth_gauge_series(rng, n, nan_p=0.15)
        bounds = [0] + cuts + [n]
        chunks = []
        for a, b in zip(bounds[:-1], bounds[1:]):
            if a < b:
                chunks.append([(int(t), float(v)) for t, v in zip(ts[a:b], vs[a:b])])
        series.append(chunks)
    st = build_store(fdb, series, kind=fdb.COL_COUNTER if fid <= 2 else None)
    start = 100000 + int(rng.integers(0, 50)) * 15000
    step = int(rng.choice([5000, 15000, 60000]))
    window = int(rng.choice([60000, 300000, 1800000]))
    nw = int(rng.integers(2, 120))
    q = fdb.make_query(start, step, start + (nw - 1) * step, window, fid)
    q.param = 0.8 if fid in (16, 17) else 600.0
    ns = st.num_series
    want = oracle.query_exec(st.view(), q, ns, nw, nthreads=4)
    got = np.empty(ns * nw, dtype=np.float64)
    engine.query(engine.upload(st), q, out=got)
    w = np.asarray(want); g = np.asarray(got)
    both_nan = np.isnan(g) & np.isnan(w)
    diff = np.where(both_nan, 0.0, np.abs(g - w) / np.maximum(np.abs(w), 1e-300))
    diff = np.where(np.isnan(diff), np.inf, diff)
    bad = np.argwhere(diff > 1e-9)
    print(f"trial {trial}: fid={fid} n={n} cuts={cuts} ns={ns} nw={nw} "
          f"start={start} step={step} win={window} bad={len(bad)}")
    for idx in bad[:8]:
        i = int(idx[0]); sid, wi = divmod(i, nw)
        print(f"  sid={sid} w={wi} got={g[i]!r} want={w[i]!r}")
