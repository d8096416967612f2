// Host-side chunk store + encoder: builds the reference's frozen BinaryVector chunk
// format from raw (timestamp, value) samples.
//
// Restates (semantics, not code) the reference ingest/encode path:
//   TimeSeriesPartition.ingest → switchBuffers(encode=true) → optimizeAll
//     (core/.../memstore/TimeSeriesPartition.scala:130)
//   timestamps: DeltaDeltaVector.fromLongVector(approxConst=true)
//     (core/.../format/vectors/LongBinaryVector.scala:333-341,
//      DeltaDeltaVector.scala:63-135)
//   doubles: DDV if all integral else raw f64; counter drop bit
//     (core/.../format/vectors/DoubleVector.scala:86-96,457-476)
//   NibblePack primitives (core/.../format/NibblePack.scala:16-183)
//
// This file is product host code (the builder feeds the GPU engine); the query-time
// DECODE restatement lives in oracle/ and is test-only.

#include <cstdint>
#include <cstring>
#include <cmath>
#include <cstdio>
#include <cstdarg>
#include <vector>
#include <string>
#include <limits>

#include "chunk_format.h"
#include "../../include/filodb_amd.h"

// ---------------------------------------------------------------------------
// error handling
// ---------------------------------------------------------------------------
static thread_local std::string g_last_error;
extern "C" const char* fdb_last_error(void) { return g_last_error.c_str(); }
void fdb_set_error(const char* fmt, ...) {
  char buf[512];
  va_list ap; va_start(ap, fmt);
  vsnprintf(buf, sizeof buf, fmt, ap);
  va_end(ap);
  g_last_error = buf;
}

// ---------------------------------------------------------------------------
// little-endian byte emit helpers (x86/amd64 host: plain stores)
// ---------------------------------------------------------------------------
using bytes = std::vector<uint8_t>;
static inline void put_u8(bytes& b, size_t off, uint8_t v)  { if (off >= b.size()) b.resize(off + 1); b[off] = v; }
static inline void put_u16(bytes& b, size_t off, uint16_t v){ if (off + 2 > b.size()) b.resize(off + 2); memcpy(&b[off], &v, 2); }
static inline void put_u32(bytes& b, size_t off, uint32_t v){ if (off + 4 > b.size()) b.resize(off + 4); memcpy(&b[off], &v, 4); }
static inline void put_u64(bytes& b, size_t off, uint64_t v){ if (off + 8 > b.size()) b.resize(off + 8); memcpy(&b[off], &v, 8); }

// ---------------------------------------------------------------------------
// NibblePack (NibblePack.scala:108-183 pack8/packUniversal; :16-98 outer packers)
// ---------------------------------------------------------------------------
// Packs 8 longs; returns new write position.
static size_t np_pack8(const int64_t in[8], bytes& buf, size_t pos) {
  int bitmask = 0;
  for (int i = 0; i < 8; i++) if (in[i] != 0) bitmask |= 1 << i;
  put_u8(buf, pos++, (uint8_t)bitmask);
  if (bitmask == 0) return pos;

  int minLeading = 64, minTrailing = 64;
  for (int i = 0; i < 8; i++) {
    uint64_t u = (uint64_t)in[i];
    // numberOfLeading/TrailingZeros(0) == 64 in Java
    int lz = u ? __builtin_clzll(u) : 64;
    int tz = u ? __builtin_ctzll(u) : 64;
    if (lz < minLeading)  minLeading  = lz;
    if (tz < minTrailing) minTrailing = tz;
  }
  int trailingNibbles = minTrailing / 4;
  int numNibbles = 16 - (minLeading / 4) - trailingNibbles;
  put_u8(buf, pos++, (uint8_t)(((numNibbles - 1) << 4) | trailingNibbles));

  // packUniversal (NibblePack.scala:143-183)
  int trailingShift = trailingNibbles * 4;
  int numBits = numNibbles * 4;
  uint64_t outWord = 0;
  int bitCursor = 0;
  for (int i = 0; i < 8; i++) {
    if (in[i] != 0) {
      int remaining = 64 - bitCursor;
      uint64_t shifted = (uint64_t)in[i] >> trailingShift;
      outWord |= (bitCursor == 64 ? 0 : shifted << bitCursor);
      if (remaining <= numBits) {
        put_u64(buf, pos, outWord); pos += 8;
        outWord = (remaining < numBits) ? (shifted >> remaining) : 0;
      }
      bitCursor = (bitCursor + numBits) % 64;
    }
  }
  if (bitCursor > 0) {
    // write remainder word but advance only the needed bytes
    if (pos + 8 > buf.size()) buf.resize(pos + 8);
    memcpy(&buf[pos], &outWord, 8);
    pos += (bitCursor + 7) / 8;
    buf.resize(pos);
  }
  return pos;
}

// packDelta (NibblePack.scala:37-63): positive nondecreasing longs as deltas.
static size_t np_pack_delta(const int64_t* input, int n, bytes& buf, size_t pos) {
  int64_t tmp[8] = {0};
  int64_t last = 0;
  int i = 0;
  for (; i < n; i++) {
    int64_t delta = (input[i] >= last) ? input[i] - last : 0;
    last = input[i];
    tmp[i % 8] = delta;
    if (i % 8 == 7) pos = np_pack8(tmp, buf, pos);
  }
  if (i % 8 != 0) {
    for (int j = i % 8; j < 8; j++) tmp[j] = 0;
    pos = np_pack8(tmp, buf, pos);
  }
  return pos;
}

// packDoubles (NibblePack.scala:73-98): first double raw, rest XOR-encoded.
static size_t np_pack_doubles(const double* input, int n, bytes& buf, size_t pos) {
  uint64_t first; memcpy(&first, &input[0], 8);
  put_u64(buf, pos, first); pos += 8;
  int64_t tmp[8] = {0};
  uint64_t last = first;
  int i = 0;
  for (; i < n - 1; i++) {
    uint64_t bits; memcpy(&bits, &input[i + 1], 8);
    tmp[i % 8] = (int64_t)(bits ^ last);
    last = bits;
    if (i % 8 == 7) pos = np_pack8(tmp, buf, pos);
  }
  if (i % 8 != 0) {
    for (int j = i % 8; j < 8; j++) tmp[j] = 0;
    pos = np_pack8(tmp, buf, pos);
  }
  return pos;
}

extern "C" int32_t fdb_nibblepack_pack8(const int64_t in[8], uint8_t* out, int32_t outcap) {
  bytes b;
  size_t n = np_pack8(in, b, 0);
  if ((int32_t)n > outcap) { fdb_set_error("output buffer too small (%zu needed)", n); return FDB_ERR_BADARG; }
  memcpy(out, b.data(), n);
  return (int32_t)n;
}
extern "C" int32_t fdb_nibblepack_pack_delta(const int64_t* in, int32_t n, uint8_t* out, int32_t outcap) {
  bytes b;
  size_t len = np_pack_delta(in, n, b, 0);
  if ((int32_t)len > outcap) { fdb_set_error("output buffer too small"); return FDB_ERR_BADARG; }
  memcpy(out, b.data(), len);
  return (int32_t)len;
}
extern "C" int32_t fdb_nibblepack_pack_doubles(const double* in, int32_t n, uint8_t* out, int32_t outcap) {
  if (n <= 0) { fdb_set_error("packDoubles requires n>0"); return FDB_ERR_BADARG; }
  bytes b;
  size_t len = np_pack_doubles(in, n, b, 0);
  if ((int32_t)len > outcap) { fdb_set_error("output buffer too small"); return FDB_ERR_BADARG; }
  memcpy(out, b.data(), len);
  return (int32_t)len;
}

// unpackDoubleXOR (NibblePack.scala:360-394): product-side host decoder for
// Gorilla-XOR packed double streams (first double raw, rest XOR-chained
// through 8-value groups).
extern "C" int32_t fdb_nibblepack_unpack8(const uint8_t* in, int32_t inlen, int64_t out[8],
                                          int32_t* consumed);
extern "C" int32_t fdb_nibblepack_unpack_doubles(const uint8_t* in, int32_t inlen,
                                                 double* out, int32_t n) {
  if (n < 1 || inlen < 8) { fdb_set_error("stream too short"); return FDB_ERR_BADARG; }
  uint64_t last; memcpy(&last, in, 8);
  memcpy(&out[0], &last, 8);
  int32_t pos = 8, i = 1;
  while (i < n) {
    int64_t grp[8]; int32_t consumed;
    if (fdb_nibblepack_unpack8(in + pos, inlen - pos, grp, &consumed) != FDB_OK)
      return FDB_ERR_BADARG;
    pos += consumed;
    for (int k = 0; k < 8 && i < n; k++, i++) {
      last ^= (uint64_t)grp[k];
      memcpy(&out[i], &last, 8);
    }
  }
  return FDB_OK;
}

// unpack8 (NibblePack.scala:395-447). Returns FDB_OK and *consumed, or error.
extern "C" int32_t fdb_nibblepack_unpack8(const uint8_t* in, int32_t inlen, int64_t out[8],
                                          int32_t* consumed) {
  if (inlen < 1) { fdb_set_error("input too short"); return FDB_ERR_BADARG; }
  uint8_t nonzeroMask = in[0];
  if (nonzeroMask == 0) {
    for (int i = 0; i < 8; i++) out[i] = 0;
    *consumed = 1;
    return FDB_OK;
  }
  if (inlen < 2) { fdb_set_error("input too short"); return FDB_ERR_BADARG; }
  int numNibblesU8 = in[1] & 0xff;
  int numBits = ((numNibblesU8 >> 4) + 1) * 4;
  int trailingZeroes = (numNibblesU8 & 0x0f) * 4;
  int totalBytes = 2 + (numBits * __builtin_popcount(nonzeroMask) + 7) / 8;
  uint64_t mask = (numBits >= 64) ? ~0ULL : ((1ULL << numBits) - 1);
  int bufIndex = 2;
  int bitCursor = 0;

  // readLong with zero-padding past the end (NibblePack.scala:458-470)
  auto readLong = [&](int idx) -> uint64_t {
    if (idx + 8 <= inlen) { uint64_t v; memcpy(&v, in + idx, 8); return v; }
    uint64_t v = 0;
    for (int i = 0; idx + i < inlen; i++) v |= (uint64_t)in[idx + i] << (8 * i);
    return v;
  };

  uint64_t inWord = readLong(bufIndex);
  bufIndex += 8;
  for (int bit = 0; bit < 8; bit++) {
    if (nonzeroMask & (1 << bit)) {
      int remaining = 64 - bitCursor;
      uint64_t shiftedIn = inWord >> bitCursor;
      uint64_t outWord = shiftedIn & mask;
      if (remaining <= numBits && bufIndex < totalBytes) {
        if (bufIndex < inlen) {
          inWord = readLong(bufIndex);
          bufIndex += 8;
          if (remaining < numBits) outWord |= (inWord << remaining) & mask;
        } else {
          fdb_set_error("InputTooShort");
          return FDB_ERR_BADARG;
        }
      }
      out[bit] = (int64_t)(outWord << trailingZeroes);
      bitCursor = (bitCursor + numBits) % 64;
    } else {
      out[bit] = 0;
    }
  }
  *consumed = totalBytes;
  return FDB_OK;
}

// ---------------------------------------------------------------------------
// DeltaDelta encoding (DeltaDeltaVector.scala:63-135)
// ---------------------------------------------------------------------------
// minMaxToNbitsSigned (IntBinaryVector.scala:161-177)
static void nbits_signed(int32_t mn, int32_t mx, int* nbits, bool* sign) {
  if (mn >= 0 && mx < 4)            { *nbits = 2;  *sign = false; }
  else if (mn >= 0 && mx < 16)      { *nbits = 4;  *sign = false; }
  else if (mn >= -128 && mx <= 127) { *nbits = 8;  *sign = true;  }
  else if (mn >= 0 && mx < 256)     { *nbits = 8;  *sign = false; }
  else if (mn >= -32768 && mx <= 32767) { *nbits = 16; *sign = true; }
  else if (mn >= 0 && mx < 65536)   { *nbits = 16; *sign = false; }
  else                              { *nbits = 32; *sign = true;  }
}

// Returns true and fills out if the long array is DDV-eligible; else false.
// approx_const: accept ±250 band as const (timestamps; DeltaDeltaVector.scala:46,77).
static bool encode_ddv_longs(const int64_t* in, int n, bool approx_const, bytes& out) {
  if (n <= 2) return false;                      // fromLongVector eligibility :67
  // getSlope (:112-115) — Long division truncates toward zero, same as C++
  int64_t slope64 = (in[n - 1] - in[0]) / (n - 1);
  if (slope64 >= INT32_MAX || slope64 <= INT32_MIN) return false;
  int32_t slope = (int32_t)slope64;
  // getDeltasMinMax (:118-130)
  int64_t base = in[0];
  int32_t mn = INT32_MAX, mx = INT32_MIN;
  for (int i = 1; i < n; i++) {
    base += slope;
    int64_t d = in[i] - base;
    if (d > INT32_MAX || d < INT32_MIN) return false;
    if ((int32_t)d > mx) mx = (int32_t)d;
    if ((int32_t)d < mn) mn = (int32_t)d;
  }
  int nbits; bool sign;
  nbits_signed(mn, mx, &nbits, &sign);           // getNbitsSignedFromMinMax ≤32 always

  if ((mn == 0 && mx == 0) ||
      (approx_const && mn >= -FDB_DDV_MAX_APPROX_DELTA && mx <= FDB_DDV_MAX_APPROX_DELTA)) {
    // const DDV (:89-106)
    out.assign(FDB_DDVC_BYTES, 0);
    put_u32(out, 0, 20);
    put_u32(out, 4, FDB_WF_DDV_CONST);
    put_u32(out, FDB_DDVC_OFF_NELEM, (uint32_t)n);
    put_u64(out, FDB_DDVC_OFF_INIT, (uint64_t)in[0]);
    put_u32(out, FDB_DDVC_OFF_SLOPE, (uint32_t)slope);
    return true;
  }

  // packed DDV: outer header + inner int vector (DeltaDeltaVector.scala:138-146;
  // inner header BinaryVector.scala:511-533, bit mechanics :560-580)
  out.clear();
  put_u32(out, 4, FDB_WF_DDV);
  put_u64(out, FDB_DDV_OFF_INIT, (uint64_t)in[0]);
  put_u32(out, FDB_DDV_OFF_SLOPE, (uint32_t)slope);
  size_t inner = FDB_DDV_OFF_INNER;
  put_u16(out, inner + 4, FDB_WF_INT_NOMASK);
  put_u8(out, inner + 6, (uint8_t)((nbits & FDB_NBITS_MASK) | (sign ? FDB_SIGN_MASK : 0)));

  size_t data = inner + FDB_PRIM_OFF_DATA;
  int bitShift = 0;
  size_t w = data;                 // current write byte
  base = in[0] - slope;            // expected = initValue; addData: delta = v - expected
  int64_t expected = in[0];
  for (int i = 0; i < n; i++) {
    int32_t v = (int32_t)(in[i] - expected);     // exact by minmax check above
    expected += slope;
    switch (nbits) {
      case 32: put_u32(out, w, (uint32_t)v); w += 4; break;
      case 16: put_u16(out, w, (uint16_t)(int16_t)v); w += 2; break;
      case 8:  put_u8(out, w, (uint8_t)(int8_t)v); w += 1; break;
      default: {  // 2 or 4, unsigned (IntBinaryVector.scala:84-105)
        uint8_t orig = (bitShift == 0) ? 0 : out[w];
        put_u8(out, w, (uint8_t)(orig | ((uint32_t)v << bitShift)));
        bitShift = (bitShift + nbits) % 8;
        if (bitShift == 0) w += 1;
        break;
      }
    }
  }
  size_t dataBytes = (w - data) + ((bitShift != 0) ? 1 : 0);
  put_u8(out, inner + 7, (uint8_t)bitShift);
  put_u32(out, inner + 0, (uint32_t)(4 + dataBytes));          // inner length word
  out.resize(data + dataBytes);
  put_u32(out, 0, (uint32_t)(out.size() - 4));                 // outer length word
  return true;
}

// raw 64-bit primitive vector (f64 or i64) with optional drop bit
static void encode_raw64(const void* vals, int n, bool drop, bytes& out) {
  out.clear();
  put_u32(out, 0, (uint32_t)(4 + 8 * n));
  put_u16(out, 4, FDB_WF_PRIM64);
  put_u16(out, 6, (uint16_t)((64 | FDB_SIGN_MASK) | (drop ? FDB_DROP_MASK : 0)));
  out.resize(FDB_PRIM_OFF_DATA + 8 * (size_t)n);
  memcpy(out.data() + FDB_PRIM_OFF_DATA, vals, 8 * (size_t)n);
}

// DoubleVector.optimize (DoubleVector.scala:86-96): DDV when all values integral.
static void encode_doubles(const double* vals, int n, bool drop, bytes& out) {
  bool all_integral = true;
  for (int i = 0; i < n; i++) {
    double d = vals[i];
    if (d > 9.2233720368547758e18 /* Long.MaxValue.toDouble */ ||
        d < -9.2233720368547758e18 /* Long.MinValue.toDouble */ || rint(d) != d) {
      all_integral = false; break;      // NaN: rint(NaN)!=NaN → non-integral
    }
  }
  if (all_integral && n > 2) {
    std::vector<int64_t> longs((size_t)n);
    for (int i = 0; i < n; i++) longs[(size_t)i] = (int64_t)vals[i];
    if (encode_ddv_longs(longs.data(), n, /*approx_const=*/false, out)) {
      if (drop) {                        // DoubleCounterAppender.optimize (:468-473)
        uint16_t w; memcpy(&w, out.data() + 6, 2);
        w |= FDB_DROP_MASK;              // bit 15 of u16 at +6 is unused by the
        memcpy(out.data() + 6, &w, 2);   // 32-bit wireformat word's upper half
      }
      return;
    }
  }
  encode_raw64(vals, n, drop, out);
}

// timestamps: approx-const DDV, else raw i64 (LongBinaryVector.scala:333-341)
static void encode_timestamps(const int64_t* ts, int n, bytes& out) {
  if (!encode_ddv_longs(ts, n, /*approx_const=*/true, out))
    encode_raw64(ts, n, false, out);
}

// ---------------------------------------------------------------------------
// chunk store
// ---------------------------------------------------------------------------
struct Chunk {
  uint64_t ts_off, val_off;      // into store blob (valid after seal)
  uint64_t max_off = 0, min_off = 0;   // hist companion columns (0 = absent)
  int32_t  num_rows;
  int64_t  start_time, end_time;
  bytes    ts_bytes, val_bytes;  // cleared after seal (moved into blob)
  bytes    max_bytes, min_bytes;
};

struct Series {
  int32_t group_id = 0;
  int32_t col_kind = FDB_COL_GAUGE;
  std::vector<int64_t> buf_ts;
  std::vector<double>  buf_vals;
  bool     buf_drop = false;          // DoubleCounterAppender drop flag (per chunk)
  double   buf_last = -1.7976931348623157e308;  // Double.MinValue
  // histogram column state (FDB_COL_HIST)
  std::vector<uint64_t> buf_hist;     // row-major [rows × num_buckets]
  std::vector<double> buf_max, buf_min;  // companion columns (otel max/min,
                                          // or the downsample count column)
  bool     has_mm = false;
  bool     has_sc = false;            // scalar sum+count (buf_max = counts)
  int32_t  num_buckets = 0;
  double   bucket_first = 0, bucket_mult = 0;
  std::vector<Chunk> chunks;
};

struct fdb_store {
  std::vector<Series> series;
  int32_t max_rows = FDB_DEFAULT_MAX_ROWS;
  bool sealed = false;
  bytes blob;
  std::vector<fdb_dir_entry_t> dir;
  std::vector<int32_t> series_first, series_nchunks, group_ids;
};

extern "C" fdb_store_t* fdb_store_create(int64_t expected_series) {
  auto* s = new fdb_store();
  if (expected_series > 0) s->series.reserve((size_t)expected_series);
  return s;
}
extern "C" void fdb_store_destroy(fdb_store_t* s) { delete s; }

extern "C" int32_t fdb_store_set_max_rows(fdb_store_t* s, int32_t max_rows) {
  if (max_rows < 1) { fdb_set_error("max_rows must be >= 1"); return FDB_ERR_BADARG; }
  s->max_rows = max_rows;
  return FDB_OK;
}

extern "C" int32_t fdb_store_add_series(fdb_store_t* s, int32_t group_id, int32_t col_kind) {
  if (s->sealed) { fdb_set_error("store is sealed"); return FDB_ERR_BADARG; }
  Series se;
  se.group_id = group_id;
  se.col_kind = col_kind;
  s->series.push_back(std::move(se));
  return (int32_t)s->series.size() - 1;
}

// packDelta of one histogram's cumulative bucket values (the wire format of
// BinaryHistogram.writeDelta, HistogramVector.scala:196-209)
static void hist_pack_raw(const uint64_t* vals, int nb, bytes& out) {
  std::vector<int64_t> v(nb);
  for (int b = 0; b < nb; b++) v[(size_t)b] = (int64_t)vals[b];
  out.clear();
  np_pack_delta(v.data(), nb, out, 0);
}

// pack8 stream of (bucket-consecutive deltas − section-base deltas)
// (NibblePack.DeltaSectDiffPackSink.process, NibblePack.scala:318-338)
static void hist_pack_diff(const int64_t* deltas, const int64_t* orig, int nb, bytes& out) {
  out.clear();
  size_t pos = 0;
  int64_t tmp[8];
  for (int i = 0; i < nb; i += 8) {
    int m = nb - i < 8 ? nb - i : 8;
    for (int k = 0; k < m; k++) tmp[k] = deltas[i + k] - orig[i + k];
    for (int k = m; k < 8; k++) tmp[k] = 0;
    pos = np_pack8(tmp, out, pos);
  }
}

// Encodes the buffered histogram rows as a sect-delta histogram vector
static void encode_hist_chunk(Series& se, int n, bytes& out) {
  const int nb = se.num_buckets;
  out.clear();
  put_u16(out, 4, FDB_WF_HIST_SECTDELTA);
  put_u16(out, FDB_HIST_OFF_NUMHIST, (uint16_t)n);
  put_u8(out, FDB_HIST_OFF_FMT, FDB_HIST_FMT_GEOMETRIC_DELTA);
  put_u16(out, FDB_HIST_OFF_DEFSIZE, 18);
  put_u16(out, FDB_HIST_OFF_DEF, (uint16_t)nb);
  uint64_t fb, mu;
  memcpy(&fb, &se.bucket_first, 8); memcpy(&mu, &se.bucket_mult, 8);
  put_u64(out, FDB_HIST_OFF_DEF + 2, fb);
  put_u64(out, FDB_HIST_OFF_DEF + 10, mu);

  std::vector<int64_t> lastDeltas(nb, 0), origDeltas(nb, 0), deltas(nb);
  bytes blob;
  size_t sect = 0;                        // current section header offset; 0 = none
  auto sect_bytes = [&]() { uint16_t v; memcpy(&v, &out[sect], 2); return (int)v; };
  auto sect_elems = [&]() { return (int)out[sect + 2]; };
  auto new_section = [&](int type) {
    sect = out.size();
    put_u16(out, sect, 0);
    put_u8(out, sect + 2, 0);
    put_u8(out, sect + 3, (uint8_t)type);
  };
  for (int e = 0; e < n; e++) {
    const uint64_t* row = se.buf_hist.data() + (size_t)e * nb;
    bool dropped = false;
    for (int b = 0; b < nb; b++) {
      int64_t d = (int64_t)row[b] - (b ? (int64_t)row[b - 1] : 0);
      if (d < lastDeltas[(size_t)b]) dropped = true;   // sink drop rule :321
      deltas[(size_t)b] = d;
    }
    bytes raw;
    hist_pack_raw(row, nb, raw);
    bool fresh;
    if (dropped) {                                     // TypeDrop section, raw base
      new_section(1);
      blob = raw;
      fresh = true;
    } else if (e == 0 || sect_elems() >= FDB_HIST_MAX_PER_SECTION ||
               sect_bytes() + (int)raw.size() >= 65536) {
      new_section(0);
      blob = raw;
      fresh = true;
    } else {
      hist_pack_diff(deltas.data(), origDeltas.data(), nb, blob);
      fresh = false;
      if (sect_elems() >= FDB_HIST_MAX_PER_SECTION ||
          sect_bytes() + (int)blob.size() >= 65536) {  // appendBlob's own check
        new_section(0);
        blob = raw;                                    // cannot happen after above; safety
        fresh = true;
      }
    }
    size_t at = out.size();
    put_u16(out, at, (uint16_t)blob.size());
    out.resize(at + 2 + blob.size());
    memcpy(out.data() + at + 2, blob.data(), blob.size());
    put_u16(out, sect, (uint16_t)(sect_bytes() + 2 + (int)blob.size()));
    put_u8(out, sect + 2, (uint8_t)(sect_elems() + 1));
    lastDeltas = deltas;
    if (fresh) origDeltas = deltas;
  }
  put_u32(out, 0, (uint32_t)(out.size() - 4));
}

static int32_t cut_chunk(fdb_store_t* s, Series& se) {
  int n = (int)se.buf_ts.size();
  if (n == 0) return FDB_OK;
  Chunk c;
  c.num_rows = n;
  c.start_time = se.buf_ts.front();
  c.end_time = se.buf_ts.back();
  encode_timestamps(se.buf_ts.data(), n, c.ts_bytes);
  if (se.col_kind == FDB_COL_HIST) {
    encode_hist_chunk(se, n, c.val_bytes);
    se.buf_hist.clear();
    if (se.has_mm) {
      encode_doubles(se.buf_max.data(), n, false, c.max_bytes);
      encode_doubles(se.buf_min.data(), n, false, c.min_bytes);
      se.buf_max.clear(); se.buf_min.clear();
    }
  } else {
    encode_doubles(se.buf_vals.data(), n, se.col_kind == FDB_COL_COUNTER && se.buf_drop,
                   c.val_bytes);
    if (se.has_sc) {                  // downsample count column rides max_off
      if ((int)se.buf_max.size() != n) {
        fdb_set_error("sum/count columns out of step (append_sc only on this series)");
        return FDB_ERR_BADARG;
      }
      encode_doubles(se.buf_max.data(), n, false, c.max_bytes);
      se.buf_max.clear();
    }
  }
  se.chunks.push_back(std::move(c));
  se.buf_ts.clear(); se.buf_vals.clear();
  se.buf_drop = false;
  se.buf_last = -1.7976931348623157e308;
  return FDB_OK;
}

extern "C" int32_t fdb_series_append_hist(fdb_store_t* s, int32_t sid,
                                          const int64_t* ts, const uint64_t* bucket_values,
                                          int32_t n, int32_t num_buckets,
                                          double bucket_first, double bucket_mult) {
  if (s->sealed) { fdb_set_error("store is sealed"); return FDB_ERR_BADARG; }
  if (sid < 0 || sid >= (int32_t)s->series.size()) { fdb_set_error("bad series id"); return FDB_ERR_BADARG; }
  Series& se = s->series[(size_t)sid];
  if (se.col_kind != FDB_COL_HIST) { fdb_set_error("series %d is not FDB_COL_HIST", sid); return FDB_ERR_BADARG; }
  if (num_buckets < 1 || num_buckets > 64) { fdb_set_error("num_buckets must be 1..64"); return FDB_ERR_BADARG; }
  if (se.num_buckets == 0) {
    se.num_buckets = num_buckets;
    se.bucket_first = bucket_first;
    se.bucket_mult = bucket_mult;
  } else if (se.num_buckets != num_buckets) {
    fdb_set_error("bucket scheme mismatch"); return FDB_ERR_BADARG;
  }
  for (int32_t i = 0; i < n; i++) {
    if (!se.buf_ts.empty() && ts[i] < se.buf_ts.back()) {
      fdb_set_error("timestamps must be nondecreasing"); return FDB_ERR_BADARG;
    }
    se.buf_ts.push_back(ts[i]);
    se.buf_hist.insert(se.buf_hist.end(), bucket_values + (size_t)i * num_buckets,
                       bucket_values + (size_t)(i + 1) * num_buckets);
    if ((int32_t)se.buf_ts.size() >= s->max_rows) {
      int32_t crc = cut_chunk(s, se);
      if (crc != FDB_OK) return crc;
    }
  }
  return FDB_OK;
}

// histogram rows with otel max/min companion double columns
// (SumAndMaxOverTimeFuncHD / CumulativeHistRateAndMinMaxFunction inputs,
//  AggrOverTimeFunctions.scala:612-813)
extern "C" int32_t fdb_series_append_hist_mm(fdb_store_t* s, int32_t sid,
                                             const int64_t* ts,
                                             const uint64_t* bucket_values,
                                             const double* maxs, const double* mins,
                                             int32_t n, int32_t num_buckets,
                                             double bucket_first, double bucket_mult) {
  if (sid < 0 || sid >= (int32_t)s->series.size()) { fdb_set_error("bad series id"); return FDB_ERR_BADARG; }
  Series& se = s->series[(size_t)sid];
  if (!se.buf_ts.empty() && !se.has_mm && se.col_kind == FDB_COL_HIST && !se.buf_hist.empty()) {
    fdb_set_error("series %d mixes hist rows with and without max/min", sid);
    return FDB_ERR_BADARG;
  }
  se.has_mm = true;
  // append row-by-row so the auto-cut keeps columns aligned
  for (int32_t i = 0; i < n; i++) {
    se.buf_max.push_back(maxs[i]);
    se.buf_min.push_back(mins[i]);
    int32_t rc = fdb_series_append_hist(s, sid, ts + i,
                                        bucket_values + (size_t)i * num_buckets,
                                        1, num_buckets, bucket_first, bucket_mult);
    if (rc != FDB_OK) return rc;
  }
  return FDB_OK;
}

// Downsampled scalar series: per row a pre-aggregated sum and its sample
// count (the downsample schema's avg path — AvgWithSumAndCountOverTimeFuncD,
// AggrOverTimeFunctions.scala:820-860: avg(window) = SumOverTime(sum col) /
// SumOverTime(count col)). The count column rides the chunk's max_off slot.
extern "C" int32_t fdb_series_append_sc(fdb_store_t* s, int32_t sid,
                                        const int64_t* ts, const double* sums,
                                        const double* counts, int32_t n) {
  if (sid < 0 || sid >= (int32_t)s->series.size()) { fdb_set_error("bad series id"); return FDB_ERR_BADARG; }
  Series& se = s->series[(size_t)sid];
  if (se.col_kind == FDB_COL_HIST) { fdb_set_error("append_sc needs a scalar series"); return FDB_ERR_BADARG; }
  if (!se.buf_ts.empty() && !se.has_sc) {
    fdb_set_error("series %d mixes rows with and without counts", sid);
    return FDB_ERR_BADARG;
  }
  se.has_sc = true;
  for (int32_t i = 0; i < n; i++) {      // row-wise so auto-cut stays aligned
    se.buf_max.push_back(counts[i]);
    int32_t rc = fdb_series_append(s, sid, ts + i, sums + i, 1);
    if (rc != FDB_OK) return rc;
  }
  return FDB_OK;
}

extern "C" int32_t fdb_series_append(fdb_store_t* s, int32_t sid,
                                     const int64_t* ts, const double* vals, int32_t n) {
  if (s->sealed) { fdb_set_error("store is sealed"); return FDB_ERR_BADARG; }
  if (sid < 0 || sid >= (int32_t)s->series.size()) { fdb_set_error("bad series id %d", sid); return FDB_ERR_BADARG; }
  Series& se = s->series[(size_t)sid];
  for (int32_t i = 0; i < n; i++) {
    if (!se.buf_ts.empty() && ts[i] < se.buf_ts.back()) {
      fdb_set_error("timestamps must be nondecreasing (series %d)", sid);
      return FDB_ERR_BADARG;
    }
    if (se.col_kind == FDB_COL_COUNTER) {
      // DoubleCounterAppender.addData (DoubleVector.scala:460-466)
      double v = vals[i];
      if (std::isnan(v) || v < se.buf_last) se.buf_drop = true;
      if (!std::isnan(v)) se.buf_last = v;
    }
    se.buf_ts.push_back(ts[i]);
    se.buf_vals.push_back(vals[i]);
    if ((int32_t)se.buf_ts.size() >= s->max_rows) {
      int32_t crc = cut_chunk(s, se);
      if (crc != FDB_OK) return crc;
    }
  }
  return FDB_OK;
}

extern "C" int32_t fdb_series_cut_chunk(fdb_store_t* s, int32_t sid) {
  if (sid < 0 || sid >= (int32_t)s->series.size()) { fdb_set_error("bad series id"); return FDB_ERR_BADARG; }
  return cut_chunk(s, s->series[(size_t)sid]);
}

extern "C" int32_t fdb_store_seal(fdb_store_t* s) {
  if (s->sealed) return FDB_OK;
  size_t total = 0, nchunks = 0;
  for (auto& se : s->series) {
    int32_t crc = cut_chunk(s, se);
    if (crc != FDB_OK) return crc;
    for (auto& c : se.chunks) {
      total += (c.ts_bytes.size() + 63 & ~size_t(63)) + (c.val_bytes.size() + 63 & ~size_t(63));
      total += (c.max_bytes.size() + 63 & ~size_t(63)) + (c.min_bytes.size() + 63 & ~size_t(63));
      nchunks++;
    }
  }
  s->blob.resize(total);
  s->dir.reserve(nchunks);
  s->series_first.reserve(s->series.size());
  s->series_nchunks.reserve(s->series.size());
  s->group_ids.reserve(s->series.size());
  size_t off = 0;
  for (auto& se : s->series) {
    s->series_first.push_back((int32_t)s->dir.size());
    s->series_nchunks.push_back((int32_t)se.chunks.size());
    s->group_ids.push_back(se.group_id);
    for (auto& c : se.chunks) {
      c.ts_off = off;
      memcpy(s->blob.data() + off, c.ts_bytes.data(), c.ts_bytes.size());
      off = (off + c.ts_bytes.size() + 63) & ~size_t(63);
      c.val_off = off;
      memcpy(s->blob.data() + off, c.val_bytes.data(), c.val_bytes.size());
      off = (off + c.val_bytes.size() + 63) & ~size_t(63);
      if (!c.max_bytes.empty()) {
        c.max_off = off;
        memcpy(s->blob.data() + off, c.max_bytes.data(), c.max_bytes.size());
        off = (off + c.max_bytes.size() + 63) & ~size_t(63);
      }
      if (!c.min_bytes.empty()) {
        c.min_off = off;
        memcpy(s->blob.data() + off, c.min_bytes.data(), c.min_bytes.size());
        off = (off + c.min_bytes.size() + 63) & ~size_t(63);
      }
      fdb_dir_entry_t e;
      e.ts_off = c.ts_off; e.val_off = c.val_off;
      e.max_off = c.max_off; e.min_off = c.min_off;
      e.start_time = c.start_time; e.end_time = c.end_time;
      e.num_rows = c.num_rows; e._pad = 0;
      s->dir.push_back(e);
    }
  }
  s->sealed = true;
  return FDB_OK;
}

extern "C" int32_t fdb_store_num_series(const fdb_store_t* s) { return (int32_t)s->series.size(); }
extern "C" int32_t fdb_store_is_sealed(const fdb_store_t* s) { return s->sealed ? 1 : 0; }
extern "C" int32_t fdb_series_num_chunks(const fdb_store_t* s, int32_t sid) {
  if (sid < 0 || sid >= (int32_t)s->series.size()) return FDB_ERR_BADARG;
  return (int32_t)s->series[(size_t)sid].chunks.size();
}

extern "C" int32_t fdb_chunk_get(const fdb_store_t* s, int32_t sid, int32_t ci, fdb_chunk_info_t* out) {
  if (sid < 0 || sid >= (int32_t)s->series.size()) { fdb_set_error("bad series id"); return FDB_ERR_BADARG; }
  const Series& se = s->series[(size_t)sid];
  if (ci < 0 || ci >= (int32_t)se.chunks.size()) { fdb_set_error("bad chunk idx"); return FDB_ERR_BADARG; }
  const Chunk& c = se.chunks[(size_t)ci];
  if (s->sealed) {
    out->ts_vec  = s->blob.data() + c.ts_off;
    out->val_vec = s->blob.data() + c.val_off;
    uint32_t tl, vl;
    memcpy(&tl, out->ts_vec, 4); memcpy(&vl, out->val_vec, 4);
    out->ts_vec_len = (int32_t)tl + 4;
    out->val_vec_len = (int32_t)vl + 4;
  } else {
    out->ts_vec = c.ts_bytes.data();
    out->val_vec = c.val_bytes.data();
    out->ts_vec_len = (int32_t)c.ts_bytes.size();
    out->val_vec_len = (int32_t)c.val_bytes.size();
  }
  out->num_rows = c.num_rows;
  out->start_time = c.start_time;
  out->end_time = c.end_time;
  return FDB_OK;
}

// structural check of one frozen scalar vector: the length word matches the
// buffer, the wireformat is known, and the decoded element count covers
// num_rows — so restored bytes can never drive the GPU decoders out of
// bounds. (Histogram vectors carry their own counts and are checked by the
// upload guard.)
static bool frozen_vec_ok(const uint8_t* b, int32_t len, int32_t num_rows,
                          int allow_hist) {
  if (len < 8) return false;
  uint32_t lw; memcpy(&lw, b, 4);
  if ((int64_t)lw + 4 != (int64_t)len) return false;
  uint16_t wf; memcpy(&wf, b + 4, 2);
  if (wf == FDB_WF_HIST_SECTDELTA) return allow_hist != 0;
  if (wf == FDB_WF_DDV_CONST) {
    if (len < FDB_DDVC_BYTES) return false;
    int32_t n; memcpy(&n, b + FDB_DDVC_OFF_NELEM, 4);
    return n >= num_rows;
  }
  if (wf == FDB_WF_DDV) {
    if (len < FDB_DDV_OFF_INNER + FDB_PRIM_OFF_DATA) return false;
    const uint8_t* inner = b + FDB_DDV_OFF_INNER;
    uint32_t ilw; memcpy(&ilw, inner, 4);
    if ((int64_t)FDB_DDV_OFF_INNER + 4 + ilw > (int64_t)len) return false;
    int nbits = inner[6] & FDB_NBITS_MASK;
    if (nbits != 2 && nbits != 4 && nbits != 8 && nbits != 16 && nbits != 32)
      return false;
    int bitShift = inner[7] & 0x3f;
    int64_t n = (((int64_t)ilw - 4) * 8 + (bitShift ? bitShift - 8 : 0)) / nbits;
    return n >= num_rows;
  }
  if (wf == FDB_WF_PRIM64)
    return ((int64_t)lw - 4) / 8 >= num_rows;
  return false;
}

// Paging-side restore (TimeSeriesChunksTable read path): appends a chunk whose
// frozen vector bytes come from persisted storage UNCHANGED — no re-encode, the
// same bytes the ODP reader hands to the query engine.
extern "C" int32_t fdb_store_add_encoded_chunk(fdb_store_t* s, int32_t sid,
                                               const uint8_t* ts_bytes, int32_t ts_len,
                                               const uint8_t* val_bytes, int32_t val_len,
                                               int32_t num_rows,
                                               int64_t start_time, int64_t end_time) {
  if (s->sealed) { fdb_set_error("store is sealed"); return FDB_ERR_BADARG; }
  if (sid < 0 || sid >= (int32_t)s->series.size()) { fdb_set_error("bad series id %d", sid); return FDB_ERR_BADARG; }
  if (ts_len < 8 || val_len < 8 || num_rows < 1) { fdb_set_error("bad encoded chunk"); return FDB_ERR_BADARG; }
  if (!frozen_vec_ok(ts_bytes, ts_len, num_rows, 0) ||
      !frozen_vec_ok(val_bytes, val_len, num_rows, 1)) {
    fdb_set_error("malformed frozen vector bytes (series %d)", sid);
    return FDB_ERR_BADARG;
  }
  Series& se = s->series[(size_t)sid];
  if (!se.buf_ts.empty()) { fdb_set_error("series %d has unsealed buffered rows", sid); return FDB_ERR_BADARG; }
  if (!se.chunks.empty() && start_time < se.chunks.back().end_time) {
    fdb_set_error("restored chunks must arrive time-ordered (series %d)", sid);
    return FDB_ERR_BADARG;
  }
  Chunk c;
  c.num_rows = num_rows;
  c.start_time = start_time;
  c.end_time = end_time;
  c.ts_bytes.assign(ts_bytes, ts_bytes + ts_len);
  c.val_bytes.assign(val_bytes, val_bytes + val_len);
  se.chunks.push_back(std::move(c));
  return FDB_OK;
}

// ---------------------------------------------------------------------------
// synthetic workload generator (TestTimeseriesProducer shapes; BASELINE configs)
// ---------------------------------------------------------------------------
static inline uint64_t splitmix64(uint64_t& x) {
  x += 0x9E3779B97F4A7C15ULL;
  uint64_t z = x;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}
static inline double u01(uint64_t& st) { return (double)(splitmix64(st) >> 11) * 0x1.0p-53; }
static int poisson_knuth(uint64_t& st, double lam) {
  double L = exp(-lam), p = 1.0;
  int k = 0;
  do { k++; p *= u01(st); } while (p > L);
  return k - 1;
}

extern "C" int32_t fdb_synth_generate(fdb_store_t* s, int32_t kind, int32_t n_series,
                                      int32_t n_samples, int64_t start_ts, int32_t step_ms,
                                      int32_t jitter_ms, double lam, double reset_p,
                                      int32_t n_groups, uint64_t seed) {
  if (s->sealed) { fdb_set_error("store is sealed"); return FDB_ERR_BADARG; }
  size_t base = s->series.size();
  s->series.resize(base + (size_t)n_series);
  for (int32_t i = 0; i < n_series; i++) {
    Series& se = s->series[base + i];
    se.group_id = n_groups > 0 ? (int32_t)((base + i) % n_groups) : 0;
    se.col_kind = kind;
  }
  #pragma omp parallel for schedule(static)
  for (int32_t i = 0; i < n_series; i++) {
    Series& se = s->series[base + i];
    uint64_t st = seed * 0x9E3779B97F4A7C15ULL + (uint64_t)(base + i) * 1000003ULL + 12345ULL;
    se.buf_ts.reserve(s->max_rows);
    se.buf_vals.reserve(s->max_rows);
    int64_t last_ts = 0;
    double counter = 0, walk = 0;
    for (int32_t k = 0; k < n_samples; k++) {
      int64_t jitter = jitter_ms > 0
        ? (int64_t)(splitmix64(st) % (uint64_t)(2 * jitter_ms + 1)) - jitter_ms : 0;
      int64_t ts = start_ts + (int64_t)k * step_ms + jitter;
      if (ts < last_ts) ts = last_ts;
      last_ts = ts;
      if (kind == FDB_COL_HIST) {
        // config #4 shape: 64 geometric buckets, per-interval cumulative counts
        // (FDB_SYNTH_HIST_NB overrides for perf experiments)
        static const int nb_env = [] {
          const char* v = getenv("FDB_SYNTH_HIST_NB");
          int x = v ? atoi(v) : 64;
          return (x >= 1 && x <= 64) ? x : 64;
        }();
        const int nb = nb_env;
        if (se.num_buckets == 0) {
          se.num_buckets = nb; se.bucket_first = 2.0; se.bucket_mult = 2.0;
          se.buf_hist.reserve((size_t)s->max_rows * nb);
        }
        static thread_local std::vector<uint64_t> cum_int;
        if (k == 0) { cum_int.assign(nb, 0); }
        if (u01(st) < reset_p) cum_int.assign(nb, 0);
        uint64_t row = 0;
        se.buf_ts.push_back(ts);
        size_t at = se.buf_hist.size();
        se.buf_hist.resize(at + nb);
        for (int b = 0; b < nb; b++) {
          cum_int[(size_t)b] += (uint64_t)poisson_knuth(st, lam / 8.0);
          row += cum_int[(size_t)b];
          se.buf_hist[at + (size_t)b] = row;
        }
        if ((int32_t)se.buf_ts.size() >= s->max_rows)
          (void)cut_chunk(s, se);   // synth columns are always aligned
        continue;
      }
      double v;
      if (kind == FDB_COL_COUNTER) {
        if (u01(st) < reset_p) counter = 0;          // counter reset
        counter += (double)poisson_knuth(st, lam);
        v = counter;
      } else {
        walk += (u01(st) - 0.5) * 2.0;
        v = walk + u01(st);                          // non-integral → raw f64 path
      }
      if (se.col_kind == FDB_COL_COUNTER) {
        if (std::isnan(v) || v < se.buf_last) se.buf_drop = true;
        if (!std::isnan(v)) se.buf_last = v;
      }
      se.buf_ts.push_back(ts);
      se.buf_vals.push_back(v);
      if ((int32_t)se.buf_ts.size() >= s->max_rows)
        (void)cut_chunk(s, se);     // synth columns are always aligned
    }
    cut_chunk(s, se);
  }
  return FDB_OK;
}

extern "C" int32_t fdb_store_view(const fdb_store_t* s, fdb_view_t* out) {
  if (!s->sealed) { fdb_set_error("store not sealed"); return FDB_ERR_BADARG; }
  out->blob = s->blob.data();
  out->blob_len = (int64_t)s->blob.size();
  out->dir = s->dir.data();
  out->num_chunks = (int64_t)s->dir.size();
  out->series_first = s->series_first.data();
  out->series_nchunks = s->series_nchunks.data();
  out->group_ids = s->group_ids.data();
  out->num_series = (int32_t)s->series.size();
  return FDB_OK;
}
