"""Encoder ↔ oracle-decoder parity for the frozen vector formats.

The builder encodes (TimeSeriesPartition.switchBuffers equivalent), the oracle
decodes with an independent restatement of the readers; round-trips must be
bit-exact for integer/timestamp data (north_star parity bar).
"""
import numpy as np

WF_DDV = 0x0808        # (SUBTYPE_INT_NOMASK<<8)|VECTORTYPE_DELTA2
WF_DDV_CONST = 0x0608
WF_PRIM64 = 0x0506

from conftest import build_store


def one_series_chunks(fdb, chunks, kind=None, max_rows=200):
    st = build_store(fdb, [chunks], kind=kind, max_rows=max_rows)
    return st


def test_timestamps_approx_const(fdb, oracle):
    # deltas within ±250 of the slope line → lossy const encoding
    # (DeltaDeltaVector.scala:46-47,75-78; LongBinaryVector.scala:333-341)
    ts = [100000 + 10000 * i + d for i, d in enumerate([0, 50, -100, 249, -249, 0])]
    st = one_series_chunks(fdb, [[(t, float(i)) for i, t in enumerate(ts)]])
    tsb, _, n, t0, t1 = st.chunk(0, 0)
    info = oracle.vec_info(tsb)
    assert info["wf"] == WF_DDV_CONST and info["n"] == 6
    dec = oracle.decode_longs(tsb)
    slope = (ts[-1] - ts[0]) // 5
    assert list(dec) == [ts[0] + slope * i for i in range(6)]
    assert t0 == ts[0] and t1 == ts[-1]  # directory keeps pre-encoding times


def test_timestamps_packed_exact(fdb, oracle):
    rng = np.random.default_rng(7)
    ts = (100000 + np.arange(50) * 10000 + rng.integers(-400, 400, 50)).astype(np.int64)
    ts = np.maximum.accumulate(ts)
    st = one_series_chunks(fdb, [[(int(t), float(i)) for i, t in enumerate(ts)]])
    tsb, _, _, _, _ = st.chunk(0, 0)
    info = oracle.vec_info(tsb)
    assert info["wf"] == WF_DDV  # jitter >250 forces packed (exact) encoding
    assert list(oracle.decode_longs(tsb)) == list(ts)  # bit-exact


def test_ddv_nbits_selection(fdb, oracle):
    # nbits ladder (IntBinaryVector.minMaxToNbitsSigned :161-177) through the
    # double-integral path (DoubleVector.optimize :86-96)
    for spread, _expect in [(1, 2), (7, 4), (100, 8), (20000, 16), (100000, 32)]:
        rng = np.random.default_rng(spread)
        base = np.arange(40, dtype=np.int64) * 1000
        vals = (base + rng.integers(0, spread + 1, 40)).astype(np.float64)
        ts = (100000 + np.arange(40) * 1000).astype(np.int64)
        st = fdb.ChunkStore()
        sid = st.add_series(0, fdb.COL_GAUGE)
        st.append(sid, ts, vals)
        st.seal()
        _, vab, _, _, _ = st.chunk(0, 0)
        info = oracle.vec_info(vab)
        assert info["wf"] == WF_DDV
        dec = oracle.decode_doubles(vab)
        assert np.array_equal(dec, vals)  # integral doubles decode bit-exact


def test_non_integral_raw_f64(fdb, oracle):
    vals = np.array([1.5, 2.25, 3.125, 4.0625, 5.5, 6.5], dtype=np.float64)
    ts = (100000 + np.arange(6) * 1000).astype(np.int64)
    st = fdb.ChunkStore()
    sid = st.add_series(0, fdb.COL_GAUGE)
    st.append(sid, ts, vals)
    st.seal()
    _, vab, _, _, _ = st.chunk(0, 0)
    info = oracle.vec_info(vab)
    assert info["wf"] == WF_PRIM64 and not info["dropped"]
    assert np.array_equal(oracle.decode_doubles(vab), vals)


def test_nan_forces_raw_and_counter_drop_bit(fdb, oracle):
    # NaN → non-integral → raw f64; counter appender marks drop on NaN or
    # decrease (DoubleVector.scala:457-476)
    ts = (100000 + np.arange(5) * 1000).astype(np.int64)
    vals = np.array([1.0, 2.0, np.nan, 4.0, 5.0])
    st = fdb.ChunkStore()
    sid = st.add_series(0, fdb.COL_COUNTER)
    st.append(sid, ts, vals)
    st.seal()
    _, vab, _, _, _ = st.chunk(0, 0)
    info = oracle.vec_info(vab)
    assert info["wf"] == WF_PRIM64 and info["dropped"]
    dec = oracle.decode_doubles(vab)
    assert np.array_equal(dec[~np.isnan(vals)], vals[~np.isnan(vals)])
    assert np.isnan(dec[2])


def test_counter_reset_integral_gets_ddv_with_drop(fdb, oracle):
    ts = (100000 + np.arange(6) * 1000).astype(np.int64)
    vals = np.array([10.0, 20.0, 30.0, 5.0, 15.0, 25.0])  # reset at idx 3
    st = fdb.ChunkStore()
    sid = st.add_series(0, fdb.COL_COUNTER)
    st.append(sid, ts, vals)
    st.seal()
    _, vab, _, _, _ = st.chunk(0, 0)
    info = oracle.vec_info(vab)
    assert info["wf"] == WF_DDV and info["dropped"]
    assert np.array_equal(oracle.decode_doubles(vab), vals)


def test_gauge_no_drop_bit(fdb, oracle):
    ts = (100000 + np.arange(4) * 1000).astype(np.int64)
    vals = np.array([10.5, 5.5, 20.5, 1.5])
    st = fdb.ChunkStore()
    sid = st.add_series(0, fdb.COL_GAUGE)
    st.append(sid, ts, vals)
    st.seal()
    _, vab, _, _, _ = st.chunk(0, 0)
    assert not oracle.vec_info(vab)["dropped"]


def test_binary_search_semantics(fdb, oracle):
    # binarySearch: first element >= item, bit31 when inexact
    # (LongBinaryVector.scala:145-152; DeltaDeltaVector.scala:159-188,245-253)
    rng = np.random.default_rng(3)
    for jitter in (0, 400):  # const and packed encodings
        ts = (100000 + np.arange(30) * 10000 +
              (rng.integers(-jitter, jitter + 1, 30) if jitter else np.zeros(30, np.int64)))
        ts = np.maximum.accumulate(ts).astype(np.int64)
        st = one_series_chunks(fdb, [[(int(t), float(i)) for i, t in enumerate(ts)]])
        tsb, _, _, _, _ = st.chunk(0, 0)
        enc = oracle.decode_longs(tsb)
        for item in [enc[0] - 5, enc[0], enc[7], enc[7] + 1, enc[-1], enc[-1] + 99]:
            r = oracle.binary_search(tsb, int(item))
            idx, exact = r & 0x7FFFFFFF, r >= 0
            expect_idx = int(np.searchsorted(enc, item, side="left"))
            assert idx == expect_idx, (jitter, item)
            assert exact == (expect_idx < len(enc) and enc[expect_idx] == item)


def test_multi_chunk_layout_and_view(fdb, oracle):
    chunks = [[(100000 + i * 1000, float(i)) for i in range(10)],
              [(120000 + i * 1000, float(i + 10)) for i in range(5)]]
    st = one_series_chunks(fdb, chunks)
    assert st.num_chunks(0) == 2
    v = st.view()
    assert v.num_series == 1 and v.num_chunks == 2
    _, _, n0, s0, e0 = st.chunk(0, 0)
    _, _, n1, s1, e1 = st.chunk(0, 1)
    assert (n0, s0, e0) == (10, 100000, 109000)
    assert (n1, s1, e1) == (5, 120000, 124000)


def test_max_rows_auto_cut(fdb):
    st = fdb.ChunkStore()
    st.set_max_rows(100)
    sid = st.add_series(0, fdb.COL_GAUGE)
    ts = (100000 + np.arange(250) * 1000).astype(np.int64)
    st.append(sid, ts, np.arange(250, dtype=np.float64) + 0.5)
    st.seal()
    assert st.num_chunks(sid) == 3  # 100+100+50 (maxChunksSize cut)


def test_huge_timestamp_jump_falls_back_to_raw(fdb, oracle):
    """Timestamp deltas beyond the 32-bit inner-vector range make
    DeltaDeltaVector.fromLongVector bail (DeltaDeltaVector.scala:63-85) —
    the encoder must fall back to raw i64 and stay bit-exact, and the window
    engine must still evaluate across the jump."""
    ts = np.array([100000, 115000, 130000,
                   100000 + (1 << 33), 100000 + (1 << 33) + 15000],
                  dtype=np.int64)
    vs = np.array([1.5, 2.5, 3.5, 4.5, 5.5])
    st = build_store(fdb, [[[(int(t), float(v)) for t, v in zip(ts, vs)]]])
    tsb, _, n, t0, t1 = st.chunk(0, 0)
    assert (n, t0, t1) == (5, int(ts[0]), int(ts[-1]))
    info = oracle.vec_info(tsb)
    assert info["wf"] == WF_PRIM64          # raw i64 fallback
    np.testing.assert_array_equal(oracle.decode_longs(tsb), ts)
    # windows on both sides of the jump
    q = fdb.make_query(int(ts[-1]), 15000, int(ts[-1]), 30000,
                       fdb.FN_SUM_OVER_TIME)
    out = oracle.eval_series(st.view(), 0, q, 1)
    assert out[0] == 4.5 + 5.5
    q2 = fdb.make_query(130000, 15000, 130000, 30000, fdb.FN_SUM_OVER_TIME)
    out2 = oracle.eval_series(st.view(), 0, q2, 1)
    assert out2[0] == 1.5 + 2.5 + 3.5


def test_ddv_size_goldens_from_reference_tests(fdb):
    """Frozen-vector size/type literals from the reference's own vector tests
    (LongVectorTest.scala:124-136 const DDV = 24 B; :100-109 packed DDV of
    [0,2,1,4,3] = 28 overhead + 3 data bytes at nbits 4; :172-185 decreasing
    const DDV = 24 B)."""
    import struct

    def enc(ts, vs, kind):
        st = fdb.ChunkStore()
        sid = st.add_series(0, kind)
        st.append(sid, np.asarray(ts, dtype=np.int64),
                  np.asarray(vs, dtype=np.float64))
        st.seal()
        return st.chunk(0, 0)[:2]

    # increasing 10s-spaced timestamps → DeltaDeltaConstVector, 24 bytes
    base = 1_694_700_000_000
    ts = [base + i * 10000 for i in range(51)]
    tsb, _ = enc(ts, [0.0] * 51, fdb.COL_GAUGE)
    assert len(tsb) == 24
    assert struct.unpack_from("<I", tsb, 4)[0] & 0xFFFF == 0x0608  # const wf

    # integral values [0,2,1,4,3]: slope 0, deltas in [0,4] → nbits 4,
    # ceil(5*4/8)=3 data bytes after the 28-byte DDV overhead
    _, vb = enc(ts[:5], [0.0, 2.0, 1.0, 4.0, 3.0], fdb.COL_GAUGE)
    assert len(vb) == 28 + 3
    assert struct.unpack_from("<I", vb, 4)[0] & 0xFFFF == 0x0808   # packed wf
    inner_nbits = vb[26] & 0x3F
    assert inner_nbits == 4

    # exactly-decreasing integral values → const DDV (all line deltas 0)
    _, vb2 = enc(ts, [float(10_000_000 - i * 100) for i in range(51)],
                 fdb.COL_GAUGE)
    assert len(vb2) == 24
    assert struct.unpack_from("<i", vb2, 20)[0] == -100            # slope
