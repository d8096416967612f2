"""BinaryRecord v2 ingestion containers.

Byte-level layout assertions follow the reference sources directly:
  RecordContainer.scala:15-27 (16-B header: len, version<<24, timestamp)
  RecordSchema.scala:60-75    (fixedStart 6, field offsets, hash at
                               offsets.last, variableAreaStart +4)
  RecordBuilder.scala:109-125 (schemaID u16 at +4), :182-190 (blob =
                               u16 len + bytes), :372-404 (map pairs =
                               1-B key len + key, u16 val len + val),
                               :461-478 (4-byte record alignment)
"""
import struct

import numpy as np
import pytest


def test_container_header_layout(fdb):
    b = fdb.BRv2Builder(ts_header=987654321)
    b.add(100000, 1.5, "m", {})
    c = b.finish()
    nbytes, verword = struct.unpack_from("<II", c, 0)
    assert nbytes == len(c) - 4            # length word counts bytes after it
    assert (verword >> 24) & 0xFF == 1     # RecordBuilder.Version
    assert struct.unpack_from("<q", c, 8)[0] == 987654321
    # first record at +16
    rec_len = struct.unpack_from("<i", c, 16)[0]
    assert 16 + 4 + rec_len <= len(c)


def test_record_field_layout(fdb):
    b = fdb.BRv2Builder()
    b.add(123456789, 42.25, "heap_usage", {"job": "api"}, schema_id=7)
    c = b.finish()
    rec = 16
    # +4 u16 schemaID; +6 i64 ts; +14 f64 value
    assert struct.unpack_from("<H", c, rec + 4)[0] == 7
    assert struct.unpack_from("<q", c, rec + 6)[0] == 123456789
    assert struct.unpack_from("<d", c, rec + 14)[0] == 42.25
    # +22/+26: u32 offsets (from record start) to the metric/tags blobs
    moff = struct.unpack_from("<I", c, rec + 22)[0]
    toff = struct.unpack_from("<I", c, rec + 26)[0]
    assert moff == 34                      # variableAreaStart
    mlen = struct.unpack_from("<H", c, rec + moff)[0]
    assert c[rec + moff + 2:rec + moff + 2 + mlen] == b"heap_usage"
    assert toff == moff + 2 + mlen
    # map: u16 total len; pair = 1-B key len + key + u16 val len + val
    map_len = struct.unpack_from("<H", c, rec + toff)[0]
    p = rec + toff + 2
    klen = c[p]
    assert klen < 0xC0                     # not a predefined-key code
    assert c[p + 1:p + 1 + klen] == b"job"
    vlen = struct.unpack_from("<H", c, p + 1 + klen)[0]
    assert c[p + 3 + klen:p + 3 + klen + vlen] == b"api"
    assert map_len == 1 + klen + 2 + vlen
    # record length word counts bytes after it; container is 4-byte aligned
    rec_len = struct.unpack_from("<i", c, rec)[0]
    assert rec + 4 + rec_len == rec + toff + 2 + map_len
    assert len(c) % 4 == 0


def test_records_align_and_iterate(fdb):
    b = fdb.BRv2Builder()
    for i in range(5):
        b.add(100000 + i * 15000, float(i), "m" + "x" * i, {"k": str(i)})
    c = b.finish()
    seen = [fdb.brv2_read(c, i) for i in range(5)]
    assert [s[0] for s in seen] == [100000 + i * 15000 for i in range(5)]
    assert [s[1] for s in seen] == [float(i) for i in range(5)]
    # identical part keys hash identically; different ones differ
    b2 = fdb.BRv2Builder()
    b2.add(1, 0.0, "m", {"k": "0"})
    b2.add(2, 1.0, "m", {"k": "0"})
    b2.add(3, 1.0, "m", {"k": "1"})
    c2 = b2.finish()
    h = [fdb.brv2_read(c2, i)[4] for i in range(3)]
    assert h[0] == h[1] != h[2]


def test_ingest_roundtrip_matches_direct_store(fdb, oracle):
    """Containers → ingest → seal → oracle query equals a directly-built
    store with the same samples (the ingest-side step feeding the path)."""
    rng = np.random.default_rng(17)
    n = 100
    series_samples = {}
    b = fdb.BRv2Builder()
    for s in range(6):
        ts = (100000 + np.arange(n) * 15000
              + rng.integers(-250, 251, n)).astype(np.int64)
        ts = np.maximum.accumulate(ts)
        vs = np.cumsum(rng.normal(0, 1, n)) + 0.5
        series_samples[s] = (ts, vs)
    # interleave records across series (ingestion order is arrival order)
    for i in range(n):
        for s in range(6):
            ts, vs = series_samples[s]
            b.add(int(ts[i]), float(vs[i]), "metric", {"job": f"j{s}"})
    container = b.finish()

    st = fdb.ChunkStore()
    ix = fdb.BRv2Index()
    nrec, nnew = fdb.ingest_brv2(st, ix, container, kind=fdb.COL_GAUGE)
    assert nrec == 6 * n and nnew == 6
    st.seal()

    st2 = fdb.ChunkStore()
    for s in range(6):
        sid = st2.add_series(0, fdb.COL_GAUGE)
        ts, vs = series_samples[s]
        st2.append(sid, ts, vs)
    st2.seal()

    q = fdb.make_query(100000 + 20 * 15000, 15000, 100000 + 90 * 15000,
                       300000, fdb.FN_SUM_OVER_TIME)
    nw = q.num_windows
    got = oracle.query_exec(st.view(), q, st.num_series, nw)
    want = oracle.query_exec(st2.view(), q, st2.num_series, nw)
    np.testing.assert_array_equal(got, want)


def test_ingest_across_containers_same_index(fdb):
    """Series identity persists across containers through the part-key index."""
    st = fdb.ChunkStore()
    ix = fdb.BRv2Index()
    for batch in range(3):
        b = fdb.BRv2Builder()
        for i in range(10):
            b.add(100000 + (batch * 10 + i) * 15000, float(i), "m",
                  {"job": "a"})
            b.add(100000 + (batch * 10 + i) * 15000, float(i) * 2, "m",
                  {"job": "b"})
        n, new = fdb.ingest_brv2(st, ix, b.finish())
        assert n == 20
        assert new == (2 if batch == 0 else 0)
    st.seal()
    assert st.num_series == 2


def test_bad_containers_rejected(fdb):
    st = fdb.ChunkStore()
    ix = fdb.BRv2Index()
    with pytest.raises(RuntimeError):
        fdb.ingest_brv2(st, ix, b"\x00" * 8)          # too short
    bad = bytearray(fdb.BRv2Builder().finish())
    bad[7] = 0x02                                      # wrong version byte
    with pytest.raises(RuntimeError):
        fdb.ingest_brv2(st, ix, bytes(bad))

# ---------------------------------------------------------------------------
# Cassandra chunk-table persistence (paging side).
# chunkid packing: core/.../store/package.scala:112-123 (startTimeShift=22 :16)
# info blob: ChunkSetInfo.scala:133-154 (chunkID@0, numRows@8,
#            ingestionTime@12, endTime@20), toBytes :250-254
# row shape: cassandra/.../columnstore/TimeSeriesChunksTable.scala:35-103
# ---------------------------------------------------------------------------

def _floor_mod(a, m):
    return a - (a // m) * m


def test_chunkid_packing(fdb):
    mod = 48 * 24 * 3600
    for st, it in [(0, 0), (1000000, 12345), (1694700000000, 999999999),
                   (5, -7), (1 << 40, mod - 1)]:
        cid = fdb.chunkid(st, it)
        want = ((1 << 63) ^ (st << 22) | _floor_mod(it, mod))
        want = want - (1 << 64) if want >= (1 << 63) else want   # as i64
        assert cid == want, (st, it)
        assert fdb.chunkid_start_time(cid) == st


def test_persist_row_layout(fdb):
    st = fdb.ChunkStore()
    sid = st.add_series(0, fdb.COL_GAUGE)
    ts = np.arange(100000, 100000 + 10 * 15000, 15000, dtype=np.int64)
    st.append(sid, ts, np.arange(10, dtype=np.float64))
    st.seal()
    pk = b"\x0a\x00heap_usage"
    rows = fdb.persist_series(st, sid, pk, ingestion_time=777)
    # frame: u32 pk_len + pk | i64 chunkid | u32 info_len + info(28) |
    #        u16 nchunks | (u32 len + bytes)*
    pl = struct.unpack_from("<I", rows, 0)[0]
    assert pl == len(pk) and rows[4:4 + pl] == pk
    p = 4 + pl
    cid = struct.unpack_from("<q", rows, p)[0]
    assert cid == fdb.chunkid(100000, 777)
    p += 8
    il = struct.unpack_from("<I", rows, p)[0]
    assert il == 28
    p += 4
    info_cid, nrows = struct.unpack_from("<qi", rows, p)
    ing, endt = struct.unpack_from("<qq", rows, p + 12)
    assert (info_cid, nrows, ing, endt) == (cid, 10, 777, int(ts[-1]))
    p += 28
    nc = struct.unpack_from("<H", rows, p)[0]
    assert nc == 2
    p += 2
    tl = struct.unpack_from("<I", rows, p)[0]
    # frozen vector bytes: leading u32 length word counts bytes after it
    assert struct.unpack_from("<I", rows, p + 4)[0] == tl - 4
    p += 4 + tl
    vl = struct.unpack_from("<I", rows, p)[0]
    assert struct.unpack_from("<I", rows, p + 4)[0] == vl - 4
    assert p + 4 + vl == len(rows)


def test_persist_restore_roundtrip_query(fdb, oracle):
    """store → chunk-table rows → fresh store: queries agree bit-exactly
    (frozen vector bytes travel UNCHANGED through the row blobs)."""
    rng = np.random.default_rng(23)
    st = fdb.ChunkStore()
    st.set_max_rows(120)                    # force multiple chunks per series
    n = 400
    for s in range(5):
        sid = st.add_series(0, fdb.COL_COUNTER)
        ts = (100000 + np.arange(n) * 15000
              + rng.integers(-250, 251, n)).astype(np.int64)
        ts = np.maximum.accumulate(ts)
        vs = np.cumsum(rng.exponential(2.0, n))
        st.append(sid, ts, vs)
    st.seal()

    st2 = fdb.ChunkStore()
    st2.set_max_rows(120)
    ix = fdb.BRv2Index()
    total = 0
    for s in range(5):
        rows = fdb.persist_series(st, s, b"pk-%d" % s, ingestion_time=1234)
        total += fdb.restore_rows(st2, ix, rows, kind=fdb.COL_COUNTER)
    assert total == sum(4 for _ in range(5))  # 400 rows / 120 → 4 chunks each
    st2.seal()

    q = fdb.make_query(100000 + 30 * 15000, 15000, 100000 + 390 * 15000,
                       300000, fdb.FN_RATE)
    got = oracle.query_exec(st2.view(), q, st2.num_series, q.num_windows)
    want = oracle.query_exec(st.view(), q, st.num_series, q.num_windows)
    np.testing.assert_array_equal(got, want)


def test_restore_series_identity_and_order(fdb):
    """Same partkey across restore calls lands in the same series; stream
    interleaving is fine; out-of-order chunk times are rejected."""
    st = fdb.ChunkStore()
    a = st.add_series(0, fdb.COL_GAUGE)
    b = st.add_series(0, fdb.COL_GAUGE)
    ts = np.arange(100000, 100000 + 50 * 15000, 15000, dtype=np.int64)
    st.append(a, ts, np.ones(50))
    st.append(b, ts, np.full(50, 2.0))
    st.seal()
    ra = fdb.persist_series(st, a, b"A")
    rb = fdb.persist_series(st, b, b"B")
    st2 = fdb.ChunkStore()
    ix = fdb.BRv2Index()
    fdb.restore_rows(st2, ix, ra)
    fdb.restore_rows(st2, ix, rb)
    fdb.restore_rows(st2, ix, b"")          # empty stream is a no-op
    assert st2.num_series == 2
    with pytest.raises(RuntimeError):       # same partkey, earlier chunk again
        fdb.restore_rows(st2, ix, ra)
    with pytest.raises(RuntimeError):       # truncated stream
        fdb.restore_rows(st2, ix, ra[:10])


@pytest.mark.gpu
def test_ingest_to_gpu_query(fdb, oracle):
    """Containers → ingest → seal → upload → GPU engine query equals the
    oracle on the same store (the full ingest-to-scan path end to end)."""
    rng = np.random.default_rng(31)
    n = 240
    b = fdb.BRv2Builder()
    series_vals = []
    for s in range(24):
        ts = (100000 + np.arange(n) * 15000
              + rng.integers(-250, 251, n)).astype(np.int64)
        ts = np.maximum.accumulate(ts)
        vs = np.cumsum(rng.exponential(2.0, n))
        series_vals.append((ts, vs))
        for t, v in zip(ts, vs):
            b.add(int(t), float(v), "req_total", {"job": f"j{s}"})
    st = fdb.ChunkStore()
    ix = fdb.BRv2Index()
    fdb.ingest_brv2(st, ix, b.finish(), kind=fdb.COL_COUNTER)
    st.seal()
    eng = fdb.Engine(0)
    q = fdb.make_query(100000 + 20 * 15000, 15000, 100000 + 239 * 15000,
                       300000, fdb.FN_RATE)
    nw = q.num_windows
    want = oracle.query_exec(st.view(), q, st.num_series, nw)
    got = np.empty(st.num_series * nw, dtype=np.float64)
    eng.query(eng.upload(st), q, out=got)
    np.testing.assert_allclose(got, want, rtol=1e-9, atol=1e-12,
                               equal_nan=True)


@pytest.mark.gpu
def test_restore_to_gpu_query(fdb, oracle):
    """Persist → restore → upload → GPU query equals the original store's
    GPU results exactly (frozen bytes unchanged through the row blobs)."""
    rng = np.random.default_rng(37)
    st = fdb.ChunkStore()
    st.set_max_rows(120)
    for s in range(16):
        sid = st.add_series(0, fdb.COL_COUNTER)
        ts = (100000 + np.arange(360) * 15000
              + rng.integers(-250, 251, 360)).astype(np.int64)
        ts = np.maximum.accumulate(ts)
        st.append(sid, ts, np.cumsum(rng.exponential(1.5, 360)))
    st.seal()
    st2 = fdb.ChunkStore()
    st2.set_max_rows(120)
    ix = fdb.BRv2Index()
    for s in range(16):
        fdb.restore_rows(st2, ix, fdb.persist_series(st, s, b"pk%d" % s, 42),
                         kind=fdb.COL_COUNTER)
    st2.seal()
    eng = fdb.Engine(0)
    q = fdb.make_query(100000 + 30 * 15000, 15000, 100000 + 350 * 15000,
                       300000, fdb.FN_RATE)
    nw = q.num_windows
    a = np.empty(16 * nw, dtype=np.float64)
    bb = np.empty(16 * nw, dtype=np.float64)
    eng.query(eng.upload(st), q, out=a)
    eng.query(eng.upload(st2), q, out=bb)
    np.testing.assert_array_equal(a, bb)


def test_persist_truncation_fuzz(fdb):
    """Every strict prefix of a persisted stream is rejected (no partial-row
    state leaks into the store)."""
    st = fdb.ChunkStore()
    sid = st.add_series(0, fdb.COL_GAUGE)
    ts = np.arange(100000, 100000 + 30 * 15000, 15000, dtype=np.int64)
    st.append(sid, ts, np.arange(30, dtype=np.float64))
    st.seal()
    rows = fdb.persist_series(st, sid, b"pk")
    for cut in range(1, len(rows), 7):
        st2 = fdb.ChunkStore()
        ix = fdb.BRv2Index()
        with pytest.raises(RuntimeError):
            fdb.restore_rows(st2, ix, rows[:cut])
        assert st2.num_series in (0, 1)      # no partial series beyond the key


def test_brv2_reader_truncation_fuzz(fdb):
    """Truncated containers never read out of bounds (error or clean count)."""
    b = fdb.BRv2Builder()
    for i in range(8):
        b.add(100000 + i * 15000, float(i), "metric", {"job": "a", "dc": "x"})
    c = b.finish()
    for cut in range(0, len(c), 5):
        st = fdb.ChunkStore()
        ix = fdb.BRv2Index()
        try:
            n, _ = fdb.ingest_brv2(st, ix, c[:cut])
            assert 0 <= n <= 8
        except RuntimeError:
            pass


def test_persist_requires_sealed_store(fdb):
    st = fdb.ChunkStore()
    sid = st.add_series(0, fdb.COL_GAUGE)
    st.append(sid, np.array([100000, 115000], dtype=np.int64),
              np.array([1.0, 2.0]))
    with pytest.raises(RuntimeError):      # buffered rows would be dropped
        fdb.persist_series(st, sid, b"pk")


def test_persist_roundtrip_idempotent(fdb):
    """persist(restore(persist(x))) == persist(x) byte for byte — the row
    blobs are a fixed point (frozen vectors travel unchanged and chunkids
    re-derive from the same start/ingestion times)."""
    rng = np.random.default_rng(71)
    st = fdb.ChunkStore()
    st.set_max_rows(90)
    sid = st.add_series(0, fdb.COL_COUNTER)
    ts = (100000 + np.arange(300) * 15000
          + rng.integers(-250, 251, 300)).astype(np.int64)
    ts = np.maximum.accumulate(ts)
    st.append(sid, ts, np.cumsum(rng.exponential(2.0, 300)))
    st.seal()
    rows1 = fdb.persist_series(st, sid, b"pk", ingestion_time=555)
    st2 = fdb.ChunkStore()
    st2.set_max_rows(90)
    ix = fdb.BRv2Index()
    fdb.restore_rows(st2, ix, rows1, kind=fdb.COL_COUNTER)
    st2.seal()
    rows2 = fdb.persist_series(st2, 0, b"pk", ingestion_time=555)
    assert rows2 == rows1


def test_restore_hostile_lengths(fdb):
    """Huge u32 field lengths in the stream must not wrap the bound checks
    (they used to overflow int32 into an in-bounds value)."""
    st = fdb.ChunkStore()
    ix = fdb.BRv2Index()
    hostile = struct.pack("<I", 0xFFFFFFF0) + b"\x00" * 64
    with pytest.raises(RuntimeError):
        fdb.restore_rows(st, ix, hostile)
    st2 = fdb.ChunkStore()
    sid = st2.add_series(0, fdb.COL_GAUGE)
    ts = np.arange(100000, 100000 + 20 * 15000, 15000, dtype=np.int64)
    st2.append(sid, ts, np.arange(20, dtype=np.float64))
    st2.seal()
    rows = bytearray(fdb.persist_series(st2, sid, b"pk"))
    # corrupt the ts-blob length to a huge value
    pl = struct.unpack_from("<I", rows, 0)[0]
    off = 4 + pl + 8 + 4 + 28 + 2
    struct.pack_into("<I", rows, off, 0xFFFFFF00)
    with pytest.raises(RuntimeError):
        fdb.restore_rows(fdb.ChunkStore(), fdb.BRv2Index(), bytes(rows))


def test_brv2_read_hostile_offset(fdb):
    """A corrupted metric-offset field must not drive the part-key copy out
    of bounds."""
    b = fdb.BRv2Builder()
    b.add(100000, 1.0, "m", {"k": "v"})
    c = bytearray(b.finish())
    struct.pack_into("<I", c, 16 + 22, 0xFFFFFFF0)   # record's metric offset
    with pytest.raises(RuntimeError):
        fdb.brv2_read(bytes(c), 0)


def test_add_encoded_chunk_rejects_malformed_vectors(fdb):
    """Restored vector bytes are structurally validated before any GPU
    decoder sees them: bad length word, unknown wireformat, element count
    short of num_rows."""
    st = fdb.ChunkStore()
    sid = st.add_series(0, fdb.COL_GAUGE)
    ts = np.arange(100000, 100000 + 20 * 15000, 15000, dtype=np.int64)
    st2 = fdb.ChunkStore()
    s2 = st2.add_series(0, fdb.COL_GAUGE)
    st2.append(s2, ts, np.arange(20, dtype=np.float64))
    st2.seal()
    tb, vb, n, t0, t1 = st2.chunk(0, 0)
    fdb.add_encoded_chunk(st, sid, tb, vb, n, t0, t1)      # well-formed: OK
    bad_len = bytearray(tb)
    struct.pack_into("<I", bad_len, 0, len(tb) + 100)      # length word lies
    with pytest.raises(RuntimeError):
        fdb.add_encoded_chunk(st, sid, bytes(bad_len), vb, n, t1 + 1, t1 + 2)
    bad_wf = bytearray(tb)
    struct.pack_into("<H", bad_wf, 4, 0x1234)              # unknown wireformat
    with pytest.raises(RuntimeError):
        fdb.add_encoded_chunk(st, sid, bytes(bad_wf), vb, n, t1 + 1, t1 + 2)
    with pytest.raises(RuntimeError):                      # count < num_rows
        fdb.add_encoded_chunk(st, sid, tb, vb, n + 50, t1 + 1, t1 + 2)
