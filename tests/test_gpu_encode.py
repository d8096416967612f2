"""GPU ingest encoder parity (SURVEY §8f ingest-side GPU encode).

Byte equality with the host encoder (chunk_builder.cpp, itself pinned to the
reference's DeltaDeltaVector/IntBinaryVector/DoubleVector layouts by the
codec byte-golden tests): every scalar shape the host encoder can emit —
const DDV (approx band), packed DDV at nbits 2/4/8/16/32, raw i64/f64
fallbacks, the counter drop bit — must come out of the wave-per-chunk
GPU kernel identical bit for bit.
"""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def engine(fdb):
    return fdb.Engine(0)


def host_chunks(fdb, chunks, kind):
    """Encode via the host path: one series per chunk, explicit cuts."""
    st = fdb.ChunkStore()
    st.set_max_rows(100000)
    for ts, vs in chunks:
        sid = st.add_series(0, kind)
        st.append(sid, np.asarray(ts, dtype=np.int64),
                  np.asarray(vs, dtype=np.float64))
        st.cut_chunk(sid)
    return [st.chunk(i, 0)[:2] for i in range(len(chunks))]


def assert_match(fdb, engine, chunks, kind):
    want = host_chunks(fdb, chunks, kind)
    got = fdb.gpu_encode_chunks(engine, chunks, kind=kind)
    for c, ((wt, wv), (gt, gv)) in enumerate(zip(want, got)):
        assert gt == wt, f"chunk {c}: ts bytes differ ({len(gt)} vs {len(wt)})"
        assert gv == wv, f"chunk {c}: val bytes differ ({len(gv)} vs {len(wv)})"


def test_encode_shapes_match_host(fdb, engine):
    rng = np.random.default_rng(41)
    n = 240
    base = 100000 + np.arange(n) * 15000

    def jittered(j):
        t = (base + rng.integers(-j, j + 1, n)).astype(np.int64)
        return np.maximum.accumulate(t)

    chunks = [
        # ts: approx-const band (±250) → const DDV; vals raw f64
        (jittered(200), rng.normal(0, 1, n)),
        # ts: jitter 3000 → packed nbits16; counter DDV vals
        (jittered(3000), np.cumsum(rng.integers(0, 50, n)).astype(np.float64)),
        # large deltas → nbits 32
        (np.cumsum(rng.integers(1, 10**6, n)).astype(np.int64) + 10**12,
         np.cumsum(rng.integers(0, 10**7, n)).astype(np.float64)),
        # integral gauge with tiny range → small-nbits value vector
        (jittered(100), (np.arange(n) % 3).astype(np.float64)),
        # NaN values → raw f64
        (jittered(100),
         np.where(rng.random(n) < 0.2, np.nan, rng.normal(5, 2, n))),
        # negative integral values → signed nbits
        (jittered(100), (rng.integers(-100, 100, n)).astype(np.float64)),
    ]
    assert_match(fdb, engine, chunks, fdb.COL_GAUGE)


def test_encode_small_nbits_and_tails(fdb, engine):
    """nbits 2/4 inner vectors (both even and ragged tail lengths) — value
    streams whose line deltas stay in [0,4) / [0,16)."""
    rng = np.random.default_rng(43)
    chunks = []
    for n in (240, 241, 199, 64, 63, 7, 3):
        ts = (100000 + np.arange(n) * 15000).astype(np.int64)
        # slope-0 value line with deltas in [0,4): v = v0 + delta_i
        v2 = 100.0 + rng.integers(0, 4, n).astype(np.float64)
        v2[0] = 100.0          # delta_0 must be 0 for [0,4) at every i
        chunks.append((ts, v2))
        v4 = 100.0 + rng.integers(0, 16, n).astype(np.float64)
        v4[0] = 100.0
        chunks.append((ts, v4))
    assert_match(fdb, engine, chunks, fdb.COL_GAUGE)


def test_encode_counter_drop_bit(fdb, engine):
    rng = np.random.default_rng(47)
    n = 200
    ts = (100000 + np.arange(n) * 15000).astype(np.int64)
    up = np.cumsum(rng.integers(1, 10, n)).astype(np.float64)
    reset = up.copy()
    reset[120:] = np.cumsum(rng.integers(1, 10, n - 120))   # counter reset
    nan_mid = up.copy()
    nan_mid[50] = np.nan                                     # NaN ⇒ drop
    chunks = [(ts, up), (ts, reset), (ts, nan_mid)]
    assert_match(fdb, engine, chunks, fdb.COL_COUNTER)
    # drop bit actually set/unset as the host does
    got = fdb.gpu_encode_chunks(engine, chunks, kind=fdb.COL_COUNTER)
    import struct
    assert struct.unpack_from("<H", got[0][1], 6)[0] & 0x8000 == 0
    assert struct.unpack_from("<H", got[1][1], 6)[0] & 0x8000 != 0


def test_encode_tiny_chunks(fdb, engine):
    """n ≤ 2 is DDV-ineligible on both sides (raw i64 + raw f64)."""
    chunks = [
        (np.array([100000], dtype=np.int64), np.array([1.5])),
        (np.array([100000, 115000], dtype=np.int64), np.array([1.0, 2.0])),
    ]
    assert_match(fdb, engine, chunks, fdb.COL_GAUGE)


def test_encode_feeds_queryable_store(fdb, oracle, engine):
    """GPU-encoded bytes → add_encoded_chunk → seal → queries equal a
    host-encoded store's (the full GPU-ingest composition)."""
    rng = np.random.default_rng(53)
    n = 240
    chunks = []
    for s in range(32):
        ts = (100000 + np.arange(n) * 15000
              + rng.integers(-3000, 3001, n)).astype(np.int64)
        ts = np.maximum.accumulate(ts)
        chunks.append((ts, np.cumsum(rng.integers(0, 40, n)).astype(np.float64)))
    enc = fdb.gpu_encode_chunks(engine, chunks, kind=fdb.COL_COUNTER)
    st = fdb.ChunkStore()
    for (ts, vs), (tb, vb) in zip(chunks, enc):
        sid = st.add_series(0, fdb.COL_COUNTER)
        fdb.add_encoded_chunk(st, sid, tb, vb, len(ts), int(ts[0]), int(ts[-1]))
    st.seal()
    st2 = fdb.ChunkStore()
    st2.set_max_rows(100000)
    for ts, vs in chunks:
        sid = st2.add_series(0, fdb.COL_COUNTER)
        st2.append(sid, ts, vs)
    st2.seal()
    q = fdb.make_query(100000 + 20 * 15000, 15000, 100000 + 239 * 15000,
                       300000, fdb.FN_RATE)
    nw = q.num_windows
    a = np.empty(32 * nw, dtype=np.float64)
    b = np.empty(32 * nw, dtype=np.float64)
    engine.query(engine.upload(st), q, out=a)
    engine.query(engine.upload(st2), q, out=b)
    np.testing.assert_array_equal(a, b)


def test_encode_fuzz_vs_host(fdb, engine):
    """Randomized shapes: mixed magnitudes, NaN densities, monotone and
    near-const streams — every chunk byte-equal to the host encoder."""
    rng = np.random.default_rng(59)
    chunks = []
    for _ in range(60):
        n = int(rng.integers(1, 400))
        style = rng.integers(0, 5)
        t0 = int(rng.integers(10**5, 10**14))
        step = int(rng.choice([1000, 15000, 60000]))
        jit = int(rng.choice([0, 100, 251, 5000, 10**6]))
        ts = t0 + np.arange(n) * step
        if jit:
            ts = np.maximum.accumulate(ts + rng.integers(-jit, jit + 1, n))
        ts = ts.astype(np.int64)
        if style == 0:
            vs = rng.normal(0, 10, n)
        elif style == 1:
            vs = np.cumsum(rng.integers(0, 100, n)).astype(np.float64)
        elif style == 2:
            vs = rng.integers(-5, 5, n).astype(np.float64)
        elif style == 3:
            vs = np.where(rng.random(n) < 0.3, np.nan, rng.normal(0, 1, n))
        else:
            vs = np.full(n, float(rng.integers(0, 10)))
        chunks.append((ts, vs))
    assert_match(fdb, engine, chunks, fdb.COL_GAUGE)
    assert_match(fdb, engine, chunks, fdb.COL_COUNTER)
