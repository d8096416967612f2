import os
import sys

import numpy as np
import pytest

# torch must initialize before the native engine touches the HIP runtime, or
# torch.cuda sees no devices (observed on ROCm 7.2); importing here guarantees
# the order for any single-file pytest invocation.
import torch  # noqa: F401

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X (run via gpurun)")


@pytest.fixture(scope="session")
def fdb():
    import filodb_amd
    return filodb_amd


@pytest.fixture(scope="session")
def oracle():
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import pyclient
    return pyclient


def build_store(fdb, series, kind=None, max_rows=400, groups=None):
    """series: list of list-of-(chunk tuples) — each series is a list of chunks,
    each chunk a list of (ts, value) pairs; chunk boundaries forced explicitly
    (mirrors RawDataWindowingSpec.timeValueRVPk + addChunkToRV,
    AggrOverTimeFunctionsSpec.scala:147-171). kind defaults to COL_COUNTER —
    the fixture schema is value:double:detectDrops=true (TestData.scala:591-596)."""
    st = fdb.ChunkStore()
    st.set_max_rows(max_rows)
    for i, chunks in enumerate(series):
        gid = groups[i] if groups else 0
        sid = st.add_series(gid, fdb.COL_COUNTER if kind is None else kind)
        for ch in chunks:
            ts = np.array([t for t, _ in ch], dtype=np.int64)
            vs = np.array([v for _, v in ch], dtype=np.float64)
            st.append(sid, ts, vs)
            st.cut_chunk(sid)
    st.seal()
    return st


def synth_counter_series(rng, n, start_ts=100000, step=15000, jitter=250,
                         lam=10.0, reset_p=0.001):
    """Config #2 shape: cumulative Poisson counters with occasional resets,
    jittered timestamps (BASELINE.json configs)."""
    ts = start_ts + np.arange(n) * step + rng.integers(-jitter, jitter + 1, n)
    ts = np.maximum.accumulate(ts)  # keep nondecreasing
    inc = rng.poisson(lam, n).astype(np.float64)
    vals = np.cumsum(inc)
    resets = rng.random(n) < reset_p
    for i in np.nonzero(resets)[0]:
        vals[i:] -= vals[i]  # counter restarts at 0 from here
    return ts.astype(np.int64), vals


def synth_gauge_series(rng, n, start_ts=100000, step=15000, jitter=400, nan_p=0.0):
    """Config #3 shape: random-walk doubles (non-integral → raw f64 path).
    jitter>250 keeps the timestamp DDV non-const (exact)."""
    ts = start_ts + np.arange(n) * step + rng.integers(-jitter, jitter + 1, n)
    ts = np.maximum.accumulate(ts)
    vals = np.cumsum(rng.normal(0, 1, n)) + rng.random(n)  # non-integral
    if nan_p > 0:
        vals[rng.random(n) < nan_p] = np.nan
    return ts.astype(np.int64), vals
