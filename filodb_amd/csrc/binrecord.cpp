// BinaryRecord v2 ingestion containers (SURVEY §8f4 — the ingest-side step
// feeding the hot path). Restates the reference's on-wire layout:
//
//   RecordContainer (binaryrecord2/RecordContainer.scala:15-27,
//                    RecordBuilder.scala:641-650):
//     +0  u32 numBytes (bytes following this word)
//     +4  u32 version word: Version(=1) << 24
//     +8  i64 server timestamp
//     +16 BinaryRecords, each 4-byte aligned (RecordBuilder.align :480)
//
//   Ingestion BinaryRecord, gauge/counter shape {timestamp: Long,
//   value: Double, partKey = [metric: String, tags: Map]},
//   partitionFieldStart = 2 (RecordSchema.scala:60-75, RecordBuilder.scala:
//   109-125, 461-478):
//     +0  i32 numBytes (bytes following)
//     +4  u16 schemaID                    (partition schema ⇒ fixedStart 6)
//     +6  i64 timestamp                   (fixed field 0)
//     +14 f64 value                       (fixed field 1)
//     +22 u32 offset→metric blob          (fixed field 2, from record start)
//     +26 u32 offset→tags map blob        (fixed field 3)
//     +30 i32 partition hash              (offsets.last; hashOffset)
//     +34 variable area: metric = u16 len + bytes (UTF8StringMedium);
//         tags map = u16 total len + pairs of [UTF8StringShort key: 1-byte
//         len + bytes, or one predefined-key code byte 0xC0|num] +
//         [UTF8StringMedium value: u16 len + bytes]
//         (RecordBuilder.addMapKeyValue :372-404, addBlob :182-190)
//
// Partition identity for ingestion = the binary-comparable partition-key
// region (the var-area bytes of the part fields — RecordSchema header
// comment :29-36); series are keyed by those bytes here exactly as the
// reference compares part keys. CAVEAT (documented parity gap): the stored
// partition hash is computed with a different 32-bit hash than the JVM's
// XXHash (the hash accelerates comparisons and round-trips opaquely; its
// VALUE is not byte-compared anywhere on this path).

#include <cstdint>
#include <cstring>
#include <string>
#include <vector>
#include <unordered_map>

#include "chunk_format.h"
#include "../../include/filodb_amd.h"

void fdb_set_error(const char* fmt, ...);

// engine-internal store access (chunk_builder.cpp)
extern "C" int32_t fdb_store_add_series(fdb_store_t* s, int32_t group_id, int32_t kind);
extern "C" int32_t fdb_series_append(fdb_store_t* s, int32_t sid,
                                     const int64_t* ts, const double* vals, int32_t n);

static const int kContainerHeader = 16;
static const int kVersion = 1;

static inline void wr_u16(std::vector<uint8_t>& b, size_t off, uint16_t v) {
  if (off + 2 > b.size()) b.resize(off + 2);
  memcpy(&b[off], &v, 2);
}
static inline void wr_u32(std::vector<uint8_t>& b, size_t off, uint32_t v) {
  if (off + 4 > b.size()) b.resize(off + 4);
  memcpy(&b[off], &v, 4);
}
static inline void wr_i64(std::vector<uint8_t>& b, size_t off, int64_t v) {
  if (off + 8 > b.size()) b.resize(off + 8);
  memcpy(&b[off], &v, 8);
}
static inline uint16_t rd_u16b(const uint8_t* p) { uint16_t v; memcpy(&v, p, 2); return v; }
static inline uint32_t rd_u32b(const uint8_t* p) { uint32_t v; memcpy(&v, p, 4); return v; }
static inline int64_t rd_i64b(const uint8_t* p) { int64_t v; memcpy(&v, p, 8); return v; }

// fixed-area geometry of the gauge/counter ingestion schema
static const int kFixedStart = 6;          // after len + schemaID
static const int kOffTs = 6, kOffVal = 14, kOffMetric = 22, kOffTags = 26;
static const int kHashOffset = 30;         // offsets.last
static const int kVarAreaStart = 34;       // 4 + offsets.last

// FNV-1a stand-in for the JVM XXHash (see header caveat)
static uint32_t pk_hash(const uint8_t* p, size_t n) {
  uint32_t h = 2166136261u;
  for (size_t i = 0; i < n; i++) { h ^= p[i]; h *= 16777619u; }
  return h;
}

struct fdb_brv2_builder {
  std::vector<uint8_t> buf;
  int32_t num_records = 0;
};

extern "C" fdb_brv2_builder_t* fdb_brv2_builder_create(int64_t ts_header) {
  auto* b = new fdb_brv2_builder();
  b->buf.resize(kContainerHeader);
  wr_u32(b->buf, 0, kContainerHeader - 4);       // EmptyNumBytes
  wr_u32(b->buf, 4, (uint32_t)kVersion << 24);   // version word
  wr_i64(b->buf, 8, ts_header);
  return b;
}

extern "C" void fdb_brv2_builder_destroy(fdb_brv2_builder_t* b) { delete b; }

extern "C" int32_t fdb_brv2_add_record(fdb_brv2_builder_t* b, int64_t ts, double val,
                                       const char* metric,
                                       const char* const* tag_kv, int32_t ntags,
                                       int32_t schema_id) {
  size_t rec = b->buf.size();                      // 4-aligned by construction
  b->buf.resize(rec + kVarAreaStart);
  wr_u16(b->buf, rec + 4, (uint16_t)schema_id);
  wr_i64(b->buf, rec + kOffTs, ts);
  uint64_t dv; memcpy(&dv, &val, 8);
  wr_i64(b->buf, rec + kOffVal, (int64_t)dv);
  // metric blob (UTF8StringMedium)
  size_t mlen = strlen(metric);
  if (mlen >= 65536) { fdb_set_error("metric too long"); return FDB_ERR_BADARG; }
  size_t end = rec + kVarAreaStart;
  wr_u32(b->buf, rec + kOffMetric, (uint32_t)(end - rec));
  wr_u16(b->buf, end, (uint16_t)mlen);
  b->buf.resize(end + 2 + mlen);
  memcpy(&b->buf[end + 2], metric, mlen);
  end += 2 + mlen;
  // tags map: u16 total len + [1-byte key len + key][u16 val len + val]*
  wr_u32(b->buf, rec + kOffTags, (uint32_t)(end - rec));
  size_t map_off = end;
  wr_u16(b->buf, map_off, 0);
  end += 2;
  for (int32_t i = 0; i < ntags; i++) {
    const char* k = tag_kv[2 * i];
    const char* v = tag_kv[2 * i + 1];
    size_t kl = strlen(k), vl = strlen(v);
    if (kl >= 192) { fdb_set_error("tag key too long (>=192)"); return FDB_ERR_BADARG; }
    if (vl >= 65536) { fdb_set_error("tag value too long"); return FDB_ERR_BADARG; }
    b->buf.resize(end + 1 + kl + 2 + vl);
    b->buf[end] = (uint8_t)kl;
    memcpy(&b->buf[end + 1], k, kl);
    end += 1 + kl;
    wr_u16(b->buf, end, (uint16_t)vl);
    memcpy(&b->buf[end + 2], v, vl);
    end += 2 + vl;
  }
  wr_u16(b->buf, map_off, (uint16_t)(end - map_off - 2));
  // partition hash over the part-field var bytes (metric blob .. map end)
  size_t pk_start = rec + kVarAreaStart;
  wr_u32(b->buf, rec + kHashOffset, pk_hash(&b->buf[pk_start], end - pk_start));
  // record length word + 4-byte alignment (RecordBuilder.endRecord :461-478)
  wr_u32(b->buf, rec, (uint32_t)(end - rec - 4));
  size_t aligned = (end + 3) & ~size_t(3);
  b->buf.resize(aligned, 0);
  // container length word
  wr_u32(b->buf, 0, (uint32_t)(b->buf.size() - 4));
  b->num_records++;
  return FDB_OK;
}

extern "C" int32_t fdb_brv2_finish(fdb_brv2_builder_t* b, uint8_t* out, int32_t cap) {
  if ((int32_t)b->buf.size() > cap) {
    fdb_set_error("container needs %zu bytes", b->buf.size());
    return FDB_ERR_BADARG;
  }
  memcpy(out, b->buf.data(), b->buf.size());
  return (int32_t)b->buf.size();
}

// walks a container; cb per record. Returns record count or error.
template <typename F>
static int32_t walk_container(const uint8_t* bytes, int32_t len, F&& cb) {
  if (len < kContainerHeader) { fdb_set_error("container too short"); return FDB_ERR_BADARG; }
  int32_t nbytes = (int32_t)rd_u32b(bytes);
  if (nbytes + 4 > len) { fdb_set_error("container length %d exceeds buffer", nbytes); return FDB_ERR_BADARG; }
  int version = (int)(rd_u32b(bytes + 4) >> 24) & 0xff;
  if (version != kVersion) { fdb_set_error("unsupported BinaryRecord container version %d", version); return FDB_ERR_BADARG; }
  int32_t end = nbytes + 4;
  int32_t cur = kContainerHeader;
  int32_t count = 0;
  while (cur < end) {
    int32_t rec_len = (int32_t)rd_u32b(bytes + cur);
    if (rec_len < kVarAreaStart - 4 ||
        (int64_t)cur + 4 + (int64_t)rec_len > (int64_t)end) {
      fdb_set_error("bad record at +%d (len %d)", cur, rec_len);
      return FDB_ERR_BADARG;
    }
    int32_t rc = cb(bytes + cur, rec_len + 4, count);
    if (rc != FDB_OK) return rc;
    count++;
    cur = (cur + 4 + rec_len + 3) & ~3;
  }
  return count;
}

// test/inspection reader: extracts one record's fields + partition-key bytes
extern "C" int32_t fdb_brv2_read(const uint8_t* bytes, int32_t len, int32_t idx,
                                 int64_t* ts, double* val,
                                 uint8_t* pk_out, int32_t pk_cap, int32_t* pk_len,
                                 int32_t* schema_id, int32_t* part_hash) {
  int32_t found = FDB_ERR_BADARG;
  int32_t rc = walk_container(bytes, len, [&](const uint8_t* rec, int32_t rlen, int32_t i) {
    if (i != idx) return FDB_OK;
    if (ts) *ts = rd_i64b(rec + kOffTs);
    if (val) memcpy(val, rec + kOffVal, 8);
    if (schema_id) *schema_id = rd_u16b(rec + 4);
    if (part_hash) *part_hash = (int32_t)rd_u32b(rec + kHashOffset);
    uint32_t pk_start = rd_u32b(rec + kOffMetric);      // first part field
    uint32_t pk_end = (uint32_t)rlen;                   // var area runs to end
    if (pk_start < (uint32_t)kVarAreaStart || pk_start > pk_end) {
      // untrusted offset: must stay inside the record's var area
      fdb_set_error("bad metric offset %u in record %d", pk_start, i);
      return FDB_ERR_BADARG;
    }
    if (pk_len) *pk_len = (int32_t)(pk_end - pk_start);
    if (pk_out) {
      if ((int32_t)(pk_end - pk_start) > pk_cap) { fdb_set_error("pk buffer too small"); return FDB_ERR_BADARG; }
      memcpy(pk_out, rec + pk_start, pk_end - pk_start);
    }
    found = FDB_OK;
    return FDB_OK;
  });
  if (rc < 0) return rc;
  return found;
}

// the ingest-side step: container records → per-series appends. Series are
// keyed by the binary partition-key region (metric + tags var bytes), exactly
// the region the reference binary-compares; new part keys create series with
// the given column kind and group 0 (tag→group mapping is the caller's
// concern — series selection precedes the hot path).
struct fdb_brv2_index {
  std::unordered_map<std::string, int32_t> by_pk;
};

extern "C" fdb_brv2_index_t* fdb_brv2_index_create(void) { return new fdb_brv2_index(); }
extern "C" void fdb_brv2_index_destroy(fdb_brv2_index_t* ix) { delete ix; }

extern "C" int32_t fdb_store_ingest_brv2(fdb_store_t* s, fdb_brv2_index_t* ix,
                                         const uint8_t* bytes, int32_t len,
                                         int32_t col_kind, int32_t* out_new_series) {
  int32_t new_series = 0;
  int32_t rc = walk_container(bytes, len, [&](const uint8_t* rec, int32_t rlen, int32_t) {
    uint32_t pk_start = rd_u32b(rec + kOffMetric);
    if (pk_start > (uint32_t)rlen) { fdb_set_error("bad part-key offset"); return FDB_ERR_BADARG; }
    std::string key((const char*)rec + pk_start, (size_t)rlen - pk_start);
    auto it = ix->by_pk.find(key);
    int32_t sid;
    if (it == ix->by_pk.end()) {
      sid = fdb_store_add_series(s, 0, col_kind);
      if (sid < 0) return sid;
      ix->by_pk.emplace(std::move(key), sid);
      new_series++;
    } else {
      sid = it->second;
    }
    int64_t ts = rd_i64b(rec + kOffTs);
    double v;
    memcpy(&v, rec + kOffVal, 8);
    return fdb_series_append(s, sid, &ts, &v, 1);
  });
  if (rc < 0) return rc;
  if (out_new_series) *out_new_series = new_series;
  return rc;    // record count
}

// ---------------------------------------------------------------------------
// Cassandra chunk-table persistence (second half — the
// paging-side format). The reference persists one row per (partition,
// chunkid) with an `info` blob and a frozen list of per-column chunk blobs
// (cassandra/.../columnstore/TimeSeriesChunksTable.scala:35-103):
//
//   partition  = the partition-key bytes
//   chunkid    = (1L<<63) ^ (startTime << 22) | floorMod(ingestionTime,
//                48*24*3600)           (core/.../store/package.scala:112-123)
//   info       = the first 28 bytes of the ChunkSetInfo record
//                {chunkID i64, numRows i32, ingestionTime i64, endTime i64}
//                (ChunkSetInfo.scala:133-154, toBytes :250-254)
//   chunks     = per data column, the frozen BinaryVector bytes in schema
//                order (timestamp, value[, max, min])
//
// There is no Cassandra here, so rows are framed into a flat byte stream:
//   per row: u32 pk_len + pk | i64 chunkid | u32 info_len + info |
//            u16 nchunks | (u32 len + bytes)*
// fdb_store_restore consumes the stream back into a store with the frozen
// bytes UNCHANGED (a true round trip: the restored view is queried
// bit-identically).

extern "C" int32_t fdb_chunk_get(const fdb_store_t* s, int32_t sid, int32_t ci,
                                 fdb_chunk_info_t* out);
extern "C" int32_t fdb_series_num_chunks(const fdb_store_t* s, int32_t sid);
extern "C" int32_t fdb_store_num_series(const fdb_store_t* s);
extern "C" int32_t fdb_store_add_encoded_chunk(fdb_store_t* s, int32_t sid,
                                               const uint8_t* ts_bytes, int32_t ts_len,
                                               const uint8_t* val_bytes, int32_t val_len,
                                               int32_t num_rows,
                                               int64_t start_time, int64_t end_time);

static const int64_t kIngestMod = 48LL * 24 * 3600;   // seconds, package.scala:115
static const int kStartTimeShift = 22;

extern "C" int64_t fdb_chunkid(int64_t start_time, int64_t ingestion_time) {
  int64_t m = ingestion_time % kIngestMod;
  if (m < 0) m += kIngestMod;                          // Math.floorMod
  return (1LL << 63) ^ (start_time << kStartTimeShift) | m;
}
extern "C" int64_t fdb_chunkid_start_time(int64_t chunkid) {
  return (int64_t)(((uint64_t)((1LL << 63) ^ chunkid)) >> kStartTimeShift);
}

extern "C" int32_t fdb_store_is_sealed(const fdb_store_t* s);

extern "C" int32_t fdb_store_persist(const fdb_store_t* s, int32_t sid,
                                     const uint8_t* partkey, int32_t pk_len,
                                     int64_t ingestion_time,
                                     uint8_t* out, int32_t cap, int32_t* out_len) {
  if (!fdb_store_is_sealed(s)) {
    // unsealed series may hold buffered rows not yet cut into chunks —
    // persisting would silently drop them
    fdb_set_error("persist requires a sealed store");
    return FDB_ERR_BADARG;
  }
  int32_t nch = fdb_series_num_chunks(s, sid);
  if (nch < 0) return nch;
  std::vector<uint8_t> buf;
  auto put = [&](const void* p, size_t n) {
    size_t o = buf.size();
    buf.resize(o + n);
    memcpy(&buf[o], p, n);
  };
  for (int32_t c = 0; c < nch; c++) {
    fdb_chunk_info_t ci;
    int32_t rc = fdb_chunk_get(s, sid, c, &ci);
    if (rc != FDB_OK) return rc;
    uint32_t pl = (uint32_t)pk_len;
    put(&pl, 4);
    put(partkey, pk_len);
    int64_t cid = fdb_chunkid(ci.start_time, ingestion_time);
    put(&cid, 8);
    uint32_t il = 28;
    put(&il, 4);
    uint8_t info[28];
    memcpy(info + 0, &cid, 8);                         // OffsetChunkID
    memcpy(info + 8, &ci.num_rows, 4);                 // OffsetNumRows
    memcpy(info + 12, &ingestion_time, 8);             // OffsetIngestionTime
    memcpy(info + 20, &ci.end_time, 8);                // OffsetEndTime
    put(info, 28);
    uint16_t nc = 2;
    put(&nc, 2);
    uint32_t tl = (uint32_t)ci.ts_vec_len;
    put(&tl, 4);
    put(ci.ts_vec, ci.ts_vec_len);
    uint32_t vl = (uint32_t)ci.val_vec_len;
    put(&vl, 4);
    put(ci.val_vec, ci.val_vec_len);
  }
  if ((int32_t)buf.size() > cap) {
    fdb_set_error("persist needs %zu bytes", buf.size());
    return FDB_ERR_BADARG;
  }
  memcpy(out, buf.data(), buf.size());
  *out_len = (int32_t)buf.size();
  return nch;
}

// restores rows into the store; partition keys map to series via the index
// (new part keys create series of col_kind). The frozen chunk bytes are kept
// verbatim — the paging-side decode is the SAME device decode the query path
// already runs.
extern "C" int32_t fdb_store_restore(fdb_store_t* s, fdb_brv2_index_t* ix,
                                     const uint8_t* bytes, int32_t len,
                                     int32_t col_kind, int32_t* out_rows) {
  int32_t pos = 0, nrows = 0;
  // 64-bit bound check: field lengths come from the (untrusted) stream, so
  // a huge u32 must not wrap the 32-bit sum into an in-bounds value
  auto need = [&](int64_t n) { return n >= 0 && (int64_t)pos + n <= (int64_t)len; };
  while (pos < len) {
    if (!need(4)) { fdb_set_error("truncated row header"); return FDB_ERR_BADARG; }
    uint32_t pl = rd_u32b(bytes + pos); pos += 4;
    if (!need((int64_t)pl + 8 + 4)) { fdb_set_error("truncated partkey"); return FDB_ERR_BADARG; }
    std::string key((const char*)bytes + pos, pl); pos += pl;
    int64_t cid = rd_i64b(bytes + pos); pos += 8;
    uint32_t il = rd_u32b(bytes + pos); pos += 4;
    if (il != 28 || !need(28 + 2)) { fdb_set_error("bad info blob"); return FDB_ERR_BADARG; }
    const uint8_t* info = bytes + pos; pos += il;
    int32_t num_rows; memcpy(&num_rows, info + 8, 4);
    int64_t end_time; memcpy(&end_time, info + 20, 8);
    int64_t start_time = fdb_chunkid_start_time(cid);
    uint16_t nc = rd_u16b(bytes + pos); pos += 2;
    if (nc != 2) { fdb_set_error("expected 2 column blobs, got %d", nc); return FDB_ERR_BADARG; }
    if (!need(4)) return FDB_ERR_BADARG;
    uint32_t tl = rd_u32b(bytes + pos); pos += 4;
    if (!need((int64_t)tl + 4)) { fdb_set_error("truncated ts blob"); return FDB_ERR_BADARG; }
    const uint8_t* tsb = bytes + pos; pos += tl;
    uint32_t vl = rd_u32b(bytes + pos); pos += 4;
    if (!need((int64_t)vl)) { fdb_set_error("truncated value blob"); return FDB_ERR_BADARG; }
    const uint8_t* vab = bytes + pos; pos += vl;
    auto it = ix->by_pk.find(key);
    int32_t sid;
    if (it == ix->by_pk.end()) {
      sid = fdb_store_add_series(s, 0, col_kind);
      if (sid < 0) return sid;
      ix->by_pk.emplace(std::move(key), sid);
    } else {
      sid = it->second;
    }
    int32_t rc = fdb_store_add_encoded_chunk(s, sid, tsb, (int32_t)tl,
                                             vab, (int32_t)vl, num_rows,
                                             start_time, end_time);
    if (rc != FDB_OK) return rc;
    nrows++;
  }
  if (out_rows) *out_rows = nrows;
  return FDB_OK;
}
