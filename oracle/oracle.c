/* oracle — CPU restatement of the FiloDB chunk-scan + range-vector query path.
 *
 * TEST INFRASTRUCTURE ONLY: this library is the parity checker and
 * the bench.py cpu_baseline leg. Only tests/, __graft_entry__.smoke() and bench.py's
 * cpu_baseline may load it. It is never linked into the product library and the
 * product GPU path never calls it.
 *
 * Every routine restates, line for line in semantics, the cited reference code:
 *   decoders   core/.../format/NibblePack.scala, vectors/DeltaDeltaVector.scala,
 *              vectors/IntBinaryVector.scala, vectors/DoubleVector.scala,
 *              vectors/LongBinaryVector.scala
 *   windowing  query/.../exec/PeriodicSamplesMapper.scala:256-331 +
 *              core/.../store/ChunkSetInfo.scala:445-529
 *   functions  query/.../rangefn/RateFunctions.scala:72-111,230-322,
 *              query/.../rangefn/AggrOverTimeFunctions.scala
 *   reduce     query/.../exec/AggrOverRangeVectors.scala:320-377 + aggregator/
 *
 * Parity pinning: the reference is Scala/JVM and cannot be built in this
 * environment (no java/sbt — probed), so the oracle is pinned by
 * golden vectors copied from the reference's own test sources into tests/
 * (NibblePackTest.scala, RateFunctionsSpec.scala, AggrOverTimeFunctionsSpec.scala).
 */
#include <stdint.h>
#include <string.h>
#include <math.h>
#include <stdlib.h>

#include "../filodb_amd/csrc/chunk_format.h"
#include "../filodb_amd/csrc/tdigest_impl.h"

#ifdef _OPENMP
#include <omp.h>
#endif

#define EXPORT __attribute__((visibility("default")))

/* ---- mirrored query structs (must match include/filodb_amd.h) ------------- */
typedef struct {
  int64_t start, step, end, window;
  int32_t func_id, agg_id, num_groups, _pad;
  double param;
  double param2;
} fdb_query_t;
typedef struct {
  const uint8_t* blob;
  int64_t blob_len;
  const fdb_dir_entry_t* dir;
  int64_t num_chunks;
  const int32_t* series_first;
  const int32_t* series_nchunks;
  const int32_t* group_ids;
  int32_t num_series, _pad;
} fdb_view_t;

enum { FN_RATE=0, FN_INCREASE=1, FN_DELTA=2, FN_SUM=3, FN_COUNT=4, FN_AVG=5,
       FN_MIN=6, FN_MAX=7, FN_STDDEV=8, FN_STDVAR=9, FN_CHANGES=10, FN_LAST=12,
       FN_PRESENT=13, FN_TIMESTAMP=14, FN_ZSCORE=15,
       FN_QUANTILE=16, FN_MAD=17, FN_PREDICT_LINEAR=18,
       FN_RATE_OVER_DELTA=19, FN_HOLT_WINTERS=20 };

/* QuantileOverTimeFunction.calculateRank (AggrOverTimeFunctions.scala:400-406)
 * + the sorted linear interpolation both quantile_over_time and
 * median_absolute_deviation_over_time share (:1227-1267). vals is sorted. */
static int cmp_dbl(const void* a, const void* b) {
  double x = *(const double*)a, y = *(const double*)b;
  return x < y ? -1 : x > y ? 1 : 0;
}
static double interp_quantile(double q, const double* vals, int n) {
  double rank = q * (n - 1);
  int lower = (int)floor(rank);
  if (lower < 0) lower = 0;
  int upper = lower + 1 < n - 1 ? lower + 1 : n - 1;
  double weight = rank - floor(rank);
  return vals[lower] * (1 - weight) + vals[upper] * weight;
}
enum { AGG_NONE=0, AGG_SUM=1, AGG_COUNT=2, AGG_MIN=3, AGG_MAX=4, AGG_AVG=5,
       AGG_TOPK=6, AGG_BOTTOMK=7, AGG_STDDEV=8, AGG_STDVAR=9, AGG_GROUP=10,
       AGG_QUANTILE=11, AGG_COUNT_VALUES=12 };

static inline uint16_t rd_u16(const uint8_t* p) { uint16_t v; memcpy(&v, p, 2); return v; }
static inline uint32_t rd_u32(const uint8_t* p) { uint32_t v; memcpy(&v, p, 4); return v; }
static inline int32_t  rd_i32(const uint8_t* p) { int32_t v; memcpy(&v, p, 4); return v; }
static inline int64_t  rd_i64(const uint8_t* p) { int64_t v; memcpy(&v, p, 8); return v; }
static inline double   rd_f64(const uint8_t* p) { double v; memcpy(&v, p, 8); return v; }

/* =========================================================================
 * NibblePack unpack8 (NibblePack.scala:395-447) — for histogram work + tests
 * ========================================================================= */
EXPORT int32_t oracle_nibblepack_unpack8(const uint8_t* in, int32_t inlen,
                                         int64_t out[8], int32_t* consumed) {
  if (inlen < 1) return -1;
  uint8_t nonzeroMask = in[0];
  if (nonzeroMask == 0) {
    for (int i = 0; i < 8; i++) out[i] = 0;
    *consumed = 1;
    return 0;
  }
  if (inlen < 2) return -1;
  int numNibblesU8 = in[1] & 0xff;
  int numBits = ((numNibblesU8 >> 4) + 1) * 4;
  int trailingZeroes = (numNibblesU8 & 0x0f) * 4;
  int totalBytes = 2 + (numBits * __builtin_popcount(nonzeroMask) + 7) / 8;
  uint64_t mask = numBits >= 64 ? ~0ULL : ((1ULL << numBits) - 1);
  int bufIndex = 2, bitCursor = 0;
  uint64_t inWord = 0;
  {
    int i; uint64_t v = 0;
    if (bufIndex + 8 <= inlen) memcpy(&v, in + bufIndex, 8);
    else for (i = 0; bufIndex + i < inlen; i++) v |= (uint64_t)in[bufIndex + i] << (8 * i);
    inWord = v;
  }
  bufIndex += 8;
  for (int bit = 0; bit < 8; bit++) {
    if (nonzeroMask & (1 << bit)) {
      int remaining = 64 - bitCursor;
      uint64_t outWord = (inWord >> bitCursor) & mask;
      if (remaining <= numBits && bufIndex < totalBytes) {
        if (bufIndex < inlen) {
          uint64_t v = 0;
          if (bufIndex + 8 <= inlen) memcpy(&v, in + bufIndex, 8);
          else { for (int i = 0; bufIndex + i < inlen; i++) v |= (uint64_t)in[bufIndex + i] << (8 * i); }
          inWord = v; bufIndex += 8;
          if (remaining < numBits) outWord |= (inWord << remaining) & mask;
        } else return -1;
      }
      out[bit] = (int64_t)(outWord << trailingZeroes);
      bitCursor = (bitCursor + numBits) % 64;
    } else out[bit] = 0;
  }
  *consumed = totalBytes;
  return 0;
}

/* =========================================================================
 * vector readers
 * ========================================================================= */
typedef struct {
  const uint8_t* p;     /* vector base */
  uint16_t wf;
  int dropped;          /* bit 15 of u16 at +6 (BinaryVector.scala:519) */
  int n;                /* element count */
  /* DDV fields */
  int64_t init; int32_t slope;
  const uint8_t* idata; /* inner packed data (+20+8) */
  int nbits, sign;
} vec_t;

static void vec_open(const uint8_t* p, vec_t* v) {
  v->p = p;
  v->wf = rd_u16(p + 4);
  v->dropped = (rd_u16(p + 6) & FDB_DROP_MASK) != 0;
  if (v->wf == FDB_WF_DDV) {
    v->init = rd_i64(p + FDB_DDV_OFF_INIT);
    v->slope = rd_i32(p + FDB_DDV_OFF_SLOPE);
    const uint8_t* inner = p + FDB_DDV_OFF_INNER;
    v->nbits = inner[6] & FDB_NBITS_MASK;
    v->sign = (inner[6] & FDB_SIGN_MASK) != 0;
    v->idata = inner + FDB_PRIM_OFF_DATA;
    /* IntVectorDataReader.length (IntBinaryVector.scala:248-250) */
    int numBytes = (int)rd_u32(inner);       /* reader numBytes = getInt(addr) */
    int bitShift = inner[7] & 0x3f;
    v->n = ((numBytes - 4) * 8 + (bitShift != 0 ? bitShift - 8 : 0)) / v->nbits;
  } else if (v->wf == FDB_WF_DDV_CONST) {
    v->n = rd_i32(p + FDB_DDVC_OFF_NELEM);
    v->init = rd_i64(p + FDB_DDVC_OFF_INIT);
    v->slope = rd_i32(p + FDB_DDVC_OFF_SLOPE);
    v->idata = 0; v->nbits = 0; v->sign = 0;
  } else { /* FDB_WF_PRIM64 raw f64/i64 */
    v->n = ((int)rd_u32(p) - 4) / 8;          /* DoubleVectorDataReader.length */
    v->idata = p + FDB_PRIM_OFF_DATA;
    v->init = 0; v->slope = 0; v->nbits = 64; v->sign = 1;
  }
}

/* inner packed-int element (IntBinaryVector.scala:306-439 readers) */
static inline int64_t inner_at(const vec_t* v, int i) {
  switch (v->nbits) {
    case 32: return rd_i32(v->idata + 4 * (size_t)i);
    case 16: { int32_t x = (int16_t)rd_u16(v->idata + 2 * (size_t)i);
               return v->sign ? x : (x & 0xffff); }
    case 8:  { int32_t x = (int8_t)v->idata[i];
               return v->sign ? x : (x & 0xff); }
    case 4:  return (v->idata[i / 2] >> ((i & 1) * 4)) & 0x0f;
    case 2:  return (v->idata[i / 4] >> ((i & 3) * 2)) & 0x03;
  }
  return 0;
}

/* long element (DeltaDeltaVector.scala:153-156,241-242; LongVectorDataReader64) */
static inline int64_t lv_at(const vec_t* v, int i) {
  if (v->wf == FDB_WF_DDV) return v->init + (int64_t)v->slope * i + inner_at(v, i);
  if (v->wf == FDB_WF_DDV_CONST) return v->init + (int64_t)v->slope * i;
  return rd_i64(v->idata + 8 * (size_t)i);
}

/* binarySearch: first element >= item; bit31 set when no exact match
 * (LongBinaryVector.scala:145-152 contract; DeltaDeltaVector.scala:159-188,245-253) */
static int lv_binary_search(const vec_t* v, int64_t item) {
  if (v->wf == FDB_WF_DDV_CONST) {
    int64_t slope = v->slope;
    int guess = slope == 0 ? (item <= v->init ? 0 : v->n)
                           : (int)((item - v->init + (slope - 1)) / slope);
    if (guess < 0) return (int)0x80000000;
    if (guess >= v->n) return (int)(0x80000000u | (uint32_t)v->n);
    if (item != lv_at(v, guess)) return (int)(0x80000000u | (uint32_t)guess);
    return guess;
  }
  if (v->wf == FDB_WF_DDV) {
    int64_t slope = v->slope;
    int len = v->n;
    int elemNo = slope == 0 ? (item <= v->init ? 0 : len)
                            : (int)((item - v->init + (slope - 1)) / slope);
    if (elemNo < 0) elemNo = 0;
    if (elemNo >= len) elemNo = len - 1;
    int64_t curBase = v->init + slope * (int64_t)elemNo;
    while (elemNo >= 0 && item < curBase + inner_at(v, elemNo)) { elemNo--; curBase -= slope; }
    if (elemNo >= 0 && item == curBase + inner_at(v, elemNo)) return elemNo;
    elemNo++; curBase += slope;
    while (elemNo < len && item > curBase + inner_at(v, elemNo)) { elemNo++; curBase += slope; }
    if (elemNo < len && item == curBase + inner_at(v, elemNo)) return elemNo;
    return (int)(0x80000000u | (uint32_t)elemNo);
  }
  /* raw i64: standard first >= (LongVectorDataReader64 binarySearch semantics) */
  int lo = 0, hi = v->n;
  while (lo < hi) { int mid = (lo + hi) >> 1; if (lv_at(v, mid) < item) lo = mid + 1; else hi = mid; }
  if (lo < v->n && lv_at(v, lo) == item) return lo;
  return (int)(0x80000000u | (uint32_t)lo);
}

/* ceilingIndex: last element <= item (LongBinaryVector.scala:162-169) */
static inline int lv_ceiling(const vec_t* v, int64_t item) {
  int r = lv_binary_search(v, item);
  if (r < 0) return (r & 0x7fffffff) - 1;
  return r;
}

/* long sum → double (DeltaDeltaConstDataReader.slopeSum :265-268) */
static inline double slope_sum(int64_t init, int32_t slope, int start, int end) {
  int len = end - start + 1;
  return (double)len * (double)(init + (int64_t)start * slope)
       + (double)(((int64_t)((end - start) * len / 2)) * slope);
}

static double lv_sum(const vec_t* v, int start, int end) {
  if (v->wf == FDB_WF_DDV_CONST) return slope_sum(v->init, v->slope, start, end);
  if (v->wf == FDB_WF_DDV) {
    int64_t s = 0;
    for (int i = start; i <= end; i++) s += inner_at(v, i);
    return slope_sum(v->init, v->slope, start, end) + (double)s;
  }
  double s = 0;
  for (int i = start; i <= end; i++) s += (double)lv_at(v, i);
  return s;
}

/* long changes (DeltaDeltaVector.scala:212-228,280-289) */
static void lv_changes(const vec_t* v, int start, int end, int64_t prev, int ignorePrev,
                       int64_t* out_changes, int64_t* out_prev) {
  if (v->wf == FDB_WF_DDV_CONST) {
    int64_t firstValue = lv_at(v, start), lastValue = lv_at(v, end);
    int64_t ch = (!ignorePrev && prev != firstValue) ? 1 : 0;
    *out_changes = v->slope == 0 ? ch : (end - start) + ch;
    *out_prev = lastValue;
    return;
  }
  int64_t prevV = prev, ch = 0;
  for (int i = start; i <= end; i++) {
    int64_t cur = lv_at(v, i);
    if (i == start && ignorePrev) prevV = cur;
    if (prevV != cur) ch++;
    prevV = cur;
  }
  *out_changes = ch; *out_prev = prevV;
}

/* ---- double reader over a value vector ----------------------------------- */
static inline double dv_at(const vec_t* v, int i) {
  if (v->wf == FDB_WF_PRIM64) return rd_f64(v->idata + 8 * (size_t)i);
  return (double)lv_at(v, i);
}

/* NaN-skipping sum (DoubleVectorDataReader64.sum :234-262; DDV wrap :553-554) */
static double dv_sum(const vec_t* v, int start, int end) {
  if (v->wf != FDB_WF_PRIM64) return lv_sum(v, start, end);
  double sum = NAN;
  for (int i = start; i <= end; i++) {
    double x = rd_f64(v->idata + 8 * (size_t)i);
    if (!isnan(x)) { if (isnan(sum)) sum = 0; sum += x; }
  }
  return sum;
}

static int dv_count(const vec_t* v, int start, int end) {
  if (v->wf != FDB_WF_PRIM64) return end - start + 1;   /* DoubleLongWrap :555 */
  int c = 0;
  for (int i = start; i <= end; i++) if (!isnan(rd_f64(v->idata + 8 * (size_t)i))) c++;
  return c;
}

/* double changes (DoubleVectorDataReader64.changes :283-302;
 * DoubleLongWrapDataReader.changes :559-566) */
static void dv_changes(const vec_t* v, int start, int end, double prev,
                       double* out_changes, double* out_prev) {
  if (v->wf != FDB_WF_PRIM64) {
    int ignorePrev = isnan(prev);
    int64_t ch, pv;
    lv_changes(v, start, end, isnan(prev) ? 0 : (int64_t)prev, ignorePrev, &ch, &pv);
    *out_changes = (double)ch; *out_prev = (double)pv;
    return;
  }
  double ch = 0, prevV = prev;
  for (int i = start; i <= end; i++) {
    double x = rd_f64(v->idata + 8 * (size_t)i);
    if (!isnan(x) && prevV != x && !isnan(prevV)) ch += 1;
    prevV = x;
  }
  *out_changes = ch; *out_prev = prevV;
}

/* =========================================================================
 * counter correction (DoubleVector.scala:177-211,305-392)
 * ========================================================================= */
typedef struct {
  int has;                /* 0 = NoCorrection */
  double lastValue;
  double correction;
} corr_meta_t;

/* per-chunk corrected-state: built lazily once per (window,chunk) visit.
 * corrected[i] = nan-zeroed value + in-chunk correction (CorrectingDoubleVectorReader
 * :325-342). scratch must hold num_rows doubles. */
typedef struct {
  const vec_t* v;
  double* corrected;      /* NULL when chunk has no drop bit */
  double chunk_correction; /* _correction total */
  int owned;              /* corrected was malloc'd (chunks > 512 rows) */
} cread_t;

static void cread_init(const vec_t* v, double* scratch, cread_t* r) {
  r->v = v;
  r->chunk_correction = 0;
  r->owned = 0;
  if (!v->dropped) { r->corrected = 0; return; }
  if (v->n > 512) {            /* eval_series scratch is 512 rows per chunk */
    scratch = (double*)malloc((size_t)v->n * sizeof(double));
    r->owned = 1;
  }
  r->corrected = scratch;
  double corr = 0, last = -1.7976931348623157e308; /* Double.MinValue */
  for (int i = 0; i < v->n; i++) {
    double x = dv_at(v, i);
    if (isnan(x)) x = 0;
    if (x < last) corr += last;
    r->corrected[i] = x + corr;
    last = x;
  }
  r->chunk_correction = corr;
}

/* test helper: the corrected series of one counter chunk — the values
 * BufferableCounterCorrectionIteratorSpec pins ([3,5,7,13,2,34] ->
 * [3,5,7,13,15,47]) and CorrectingDoubleVectorReader materializes. */
EXPORT int32_t oracle_corrected_doubles(const uint8_t* vec, double* out,
                                        int32_t cap) {
  vec_t v;
  vec_open(vec, &v);
  if (v.n > cap) return -1;
  double corr = 0, last = -1.7976931348623157e308;
  for (int i = 0; i < v.n; i++) {
    double x = dv_at(&v, i);
    if (isnan(x)) x = 0;
    if (x < last) corr += last;
    out[i] = x + corr;
    last = x;
  }
  return v.n;
}

static inline double corrected_value(const cread_t* r, int n, const corr_meta_t* m) {
  double corr = m->has ? m->correction : 0;
  if (r->corrected) return r->corrected[n] + corr;
  return dv_at(r->v, n) + corr;
}

static void detect_drop_and_correction(const vec_t* v, corr_meta_t* m) {
  if (!m->has) return;
  double first = dv_at(v, 0);
  if (isnan(first) || first < m->lastValue) m->correction += m->lastValue;
}

static void update_correction(const cread_t* r, corr_meta_t* m) {
  const vec_t* v = r->v;
  double lastValue;
  if (r->corrected) {      /* CorrectingDoubleVectorReader.updateCorrection :375-391 */
    int index = v->n - 1;
    lastValue = dv_at(v, index); index--;
    while (isnan(lastValue) && index >= 0) { lastValue = dv_at(v, index); index--; }
    if (isnan(lastValue)) lastValue = 0;
    m->correction = (m->has ? m->correction : 0) + r->chunk_correction;
  } else {                 /* default updateCorrection :190-195 */
    lastValue = dv_at(v, v->n - 1);
    if (!m->has) m->correction = 0;
  }
  m->lastValue = lastValue;
  m->has = 1;
}

/* =========================================================================
 * extrapolatedRate (RateFunctions.scala:72-111)
 * ========================================================================= */
static double extrapolated_rate(int64_t windowStart, int64_t windowEnd, int numSamples,
                                int64_t t1, double v1, int64_t t2, double v2,
                                int isCounter, int isRate) {
  double durationToStart = (double)(t1 - windowStart) / 1000.0;
  double durationToEnd = (double)(windowEnd - t2) / 1000.0;
  double sampledInterval = (double)(t2 - t1) / 1000.0;
  double averageDurationBetweenSamples = sampledInterval / ((double)numSamples - 1);
  double delta = v2 - v1;
  if (isCounter && delta > 0 && v1 >= 0) {
    double durationToZero = sampledInterval * (v1 / delta);
    if (durationToZero < durationToStart) durationToStart = durationToZero;
  }
  double extrapolationThreshold = averageDurationBetweenSamples * 1.1;
  double extrapolateToInterval = sampledInterval;
  extrapolateToInterval += (durationToStart < extrapolationThreshold)
                             ? durationToStart : averageDurationBetweenSamples / 2;
  extrapolateToInterval += (durationToEnd < extrapolationThreshold)
                             ? durationToEnd : averageDurationBetweenSamples / 2;
  double scaledDelta = delta * (extrapolateToInterval / sampledInterval);
  return isRate ? (scaledDelta / (double)(windowEnd - windowStart) * 1000.0) : scaledDelta;
}

/* =========================================================================
 * per-series window evaluation (ChunkedWindowIterator.doNext,
 * PeriodicSamplesMapper.scala:293-330 + WindowedChunkIterator,
 * ChunkSetInfo.scala:467-529; function semantics DESIGN.md §3)
 * ========================================================================= */
typedef struct { double* scratch; } eval_ctx_t;

static void eval_series(const fdb_view_t* view, int sid, const fdb_query_t* q,
                        eval_ctx_t* ctx, double* out /* numWindows */) {
  int nw = (int)((q->end - q->start) / q->step) + 1;
  int first = view->series_first[sid];
  int nchunks = view->series_nchunks[sid];
  const fdb_dir_entry_t* dir = view->dir + first;
  int is_rate_family = q->func_id <= FN_DELTA;

  /* pre-open vectors once per series */
  vec_t tsv[64], vav[64];
  cread_t cr[64];
  double* scratch = ctx->scratch;
  if (nchunks > 64) nchunks = 64;  /* oracle cap; builder max_rows keeps chunks few */
  int total_rows = 0;
  for (int c = 0; c < nchunks; c++) {
    vec_open(view->blob + dir[c].ts_off, &tsv[c]);
    vec_open(view->blob + dir[c].val_off, &vav[c]);
    total_rows += dir[c].num_rows;
    if (is_rate_family) {
      cread_init(&vav[c], scratch + (size_t)c * 512, &cr[c]);
    }
  }
  /* quantile/MAD need the window's raw samples materialized */
  double* qbuf = (q->func_id == FN_QUANTILE || q->func_id == FN_MAD)
                     ? (double*)malloc((size_t)(total_rows > 0 ? total_rows : 1)
                                       * sizeof(double))
                     : NULL;

  for (int w = 0; w < nw; w++) {
    int64_t wEnd = q->start + (int64_t)w * q->step;
    int64_t wStart = wEnd - q->window;   /* inclusive-range=true (ChunkSetInfo.scala:470-473) */
    double result = NAN;

    if (is_rate_family) {
      /* ChunkedRateFunctionBase (RateFunctions.scala:230-289) +
       * CounterChunkedRangeFunction.addChunks (RangeFunction.scala:138-163) */
      corr_meta_t meta = {0, 0, 0};
      int numSamples = 0;
      int64_t lowestTime = INT64_MAX, highestTime = 0;
      double lowestValue = NAN, highestValue = NAN;
      int isCounter = q->func_id != FN_DELTA;
      for (int c = 0; c < nchunks; c++) {
        if (dir[c].end_time < wStart) continue;   /* WindowedChunkIterator drop rule */
        const vec_t* tv = &tsv[c];
        int startRow = lv_binary_search(tv, wStart) & 0x7fffffff;
        int endRow = lv_ceiling(tv, wEnd);
        if (endRow > dir[c].num_rows - 1) endRow = dir[c].num_rows - 1;
        if (isCounter) detect_drop_and_correction(&vav[c], &meta);
        if (startRow <= endRow) {
          int skip = 0;
          if (isCounter && startRow == 0 && endRow == 0 && isnan(dv_at(&vav[c], 0)))
            skip = 1;  /* single-row NaN chunk (RateFunctions.scala:253-255) */
          if (!skip) {
            int64_t st = lv_at(tv, startRow), en = lv_at(tv, endRow);
            if (st < lowestTime || en > highestTime) {
              numSamples += endRow - startRow + 1;
              if (st < lowestTime) {
                lowestTime = st;
                lowestValue = isCounter ? corrected_value(&cr[c], startRow, &meta)
                                        : dv_at(&vav[c], startRow);
              }
              if (en > highestTime) {
                highestTime = en;
                highestValue = isCounter ? corrected_value(&cr[c], endRow, &meta)
                                         : dv_at(&vav[c], endRow);
              }
            }
          }
        }
        if (isCounter) update_correction(&cr[c], &meta);
        if (dir[c].end_time >= wEnd) break;       /* add-while rule */
      }
      if (highestTime > lowestTime) {
        result = extrapolated_rate(wStart, wEnd, numSamples,
                                   lowestTime, lowestValue, highestTime, highestValue,
                                   q->func_id != FN_DELTA, q->func_id == FN_RATE);
      }
    } else {
      /* gauge family state (AggrOverTimeFunctions.scala) */
      double sum = NAN, count = NAN, sqsum = NAN, mn = NAN, mx = NAN;
      double changes = NAN, prev = NAN;
      double last_val = NAN;
      double last_sample = NAN;    /* zscore: endRow value when non-NaN */
      int64_t last_ts = -1;
      int icount = 0;
      int qn = 0, touched = 0;     /* quantile/MAD window sample buffer */
      double plX = NAN, plY = NAN, plXY = NAN, plX2 = NAN;  /* predict_linear */
      int plN = 0;
      double hwS = NAN, hwB = NAN, hwNext = NAN, hwRes = NAN;  /* holt_winters */
      for (int c = 0; c < nchunks; c++) {
        if (dir[c].end_time < wStart) continue;
        const vec_t* tv = &tsv[c];
        const vec_t* vv = &vav[c];
        int startRow = lv_binary_search(tv, wStart) & 0x7fffffff;
        int endRow = lv_ceiling(tv, wEnd);
        if (endRow > dir[c].num_rows - 1) endRow = dir[c].num_rows - 1;
        if (q->func_id == FN_LAST || q->func_id == FN_PRESENT) {
          /* LastSampleChunkedFunction.addChunks (RangeFunction.scala:599-614):
           * no startRow search; last ts <= wEnd wins if within the window.
           * PresentOverTimeChunkedFunctionD (:725-745): non-NaN -> 1; a NaN
           * stale marker steps back one row (marker-before-marker -> NaN). */
          if (endRow >= 0) {
            int64_t t = lv_at(tv, endRow);
            if (t >= wStart && t > last_ts) {
              double v = dv_at(vv, endRow);
              if (q->func_id == FN_LAST) { last_ts = t; last_val = v; }
              else if (!isnan(v)) { last_ts = t; last_val = 1; }
              else if (endRow > 0) {
                last_ts = t;
                last_val = isnan(dv_at(vv, endRow - 1)) ? NAN : 1;
              }
            }
          }
          if (dir[c].end_time >= wEnd) break;
          continue;
        }
        if (q->func_id == FN_TIMESTAMP) {
          /* TimestampChunkedFunction (RangeFunction.scala:705-723): last
           * ts <= wEnd in the chunk list, no window-start bound; seconds */
          if (endRow >= 0) {
            int64_t t = lv_at(tv, endRow);
            if (t > last_ts) { last_ts = t; last_val = (double)t / 1000.0; }
          }
          if (dir[c].end_time >= wEnd) break;
          continue;
        }
        if (startRow <= endRow) {
          switch (q->func_id) {
            case FN_SUM: case FN_RATE_OVER_DELTA: {
              double cs = dv_sum(vv, startRow, endRow);
              if (!isnan(cs) && isnan(sum)) sum = 0;
              sum += cs;                       /* :560-572 incl. NaN-poison quirk */
            } break;
            case FN_COUNT: {
              if (isnan(count)) count = 0;     /* :943-958 */
              count += dv_count(vv, startRow, endRow);
            } break;
            case FN_AVG: {
              double cs = dv_sum(vv, startRow, endRow);
              if (!isnan(cs) && isnan(sum)) sum = 0;
              sum += cs;
              icount += dv_count(vv, startRow, endRow);   /* :1004-1016 */
            } break;
            case FN_MIN: case FN_MAX: {
              for (int i = startRow; i <= endRow; i++) {
                double x = dv_at(vv, i);
                if (isnan(x)) continue;        /* QueryUtils.minIgnoreNaN */
                if (q->func_id == FN_MIN) mn = isnan(mn) || x < mn ? x : mn;
                else mx = isnan(mx) || x > mx ? x : mx;
              }
            } break;
            case FN_STDDEV: case FN_STDVAR: case FN_ZSCORE: {
              /* VarOverTimeChunkedFunctionD :1082-1115; lastSample is set
               * only when the chunk range's endRow value is non-NaN (:1103) */
              double cs = NAN, csq = NAN; int cc = 0;
              for (int i = startRow; i <= endRow; i++) {
                double x = dv_at(vv, i);
                if (!isnan(x)) {
                  if (isnan(cs)) cs = 0;
                  if (isnan(csq)) csq = 0;
                  if (i == endRow) last_sample = x;
                  cs += x; csq += x * x; cc++;
                }
              }
              if (!isnan(cs) && isnan(sum)) sum = 0;
              sum += cs;
              if (!isnan(csq) && isnan(sqsum)) sqsum = 0;
              sqsum += csq;
              icount += cc;
            } break;
            case FN_CHANGES: {
              if (isnan(changes)) changes = 0; /* :1185-1210 */
              double ch, pv;
              dv_changes(vv, startRow, endRow, prev, &ch, &pv);
              changes += ch; prev = pv;
            } break;
            case FN_PREDICT_LINEAR: {
              /* PredictLinearChunkedFunctionD (AggrOverTimeFunctions.scala:
               * 1520-1554): regression sums over x=(ts-wEnd)/1000, y=value */
              for (int i = startRow; i <= endRow; i++) {
                double y = dv_at(vv, i);
                if (isnan(y)) continue;
                double x = (double)(lv_at(tv, i) - wEnd) / 1000.0;
                if (isnan(plY)) { plY = y; plX = x; plXY = x * y; plX2 = x * x; }
                else { plY += y; plX += x; plXY += x * y; plX2 += x * x; }
                plN++;
              }
            } break;
            case FN_HOLT_WINTERS: {
              /* HoltWintersChunkedFunctionD.addTimeDoubleChunks
               * (AggrOverTimeFunctions.scala:1379-1452), restated operation
               * for operation. The reference's final it.next of every chunk
               * range reads one row PAST endRow: inside the chunk that is
               * the decoded row endRow+1; at the chunk's end it is whatever
               * memory follows the vector — undefined — modeled here as NaN.
               * The engine only accepts single-chunk series, where that
               * value provably never reaches the recurrence. */
              double sf = q->param, tf = q->param2;
              int itPos = startRow;     /* next row the iterator returns */
              int rowNum = startRow;
              if (isnan(hwS) && isnan(hwB)) {
                double s0v = NAN, b0v = NAN;
                int cur = startRow;
                while (cur <= endRow && isnan(s0v)) { s0v = dv_at(vv, itPos++); cur++; }
                while (cur <= endRow && isnan(b0v)) { b0v = dv_at(vv, itPos++); cur++; }
                hwNext = b0v;
                hwB = b0v - s0v;
                hwS = s0v;
                rowNum = cur - 1;
              } else if (isnan(hwB)) {
                double b0v = NAN;
                int cur = startRow;
                while (cur <= endRow && isnan(b0v)) { b0v = dv_at(vv, itPos++); cur++; }
                hwNext = b0v;
                hwB = b0v - hwS;
                rowNum = cur - 1;
              } else {
                itPos++;                /* continuation discards one read */
              }
              if (!isnan(hwB)) {
                while (rowNum <= endRow) {
                  if (!isnan(hwNext)) {
                    double ns = sf * hwNext + (1 - sf) * (hwS + hwB);
                    hwB = tf * (ns - hwS) + (1 - tf) * hwB;
                    hwS = ns;
                  }
                  hwNext = (itPos < dir[c].num_rows) ? dv_at(vv, itPos) : NAN;
                  itPos++;
                  rowNum++;
                }
                hwRes = hwS;
              }
            } break;
            case FN_QUANTILE: case FN_MAD: {
              /* QuantileOverTimeChunkedFunctionD (:1272-1299) /
               * MedianAbsoluteDeviationOverTimeChunkedFunctionD (:1302-1330):
               * collect the window's non-NaN samples */
              touched = 1;
              if (!(q->func_id == FN_QUANTILE && (q->param < 0 || q->param > 1)))
                for (int i = startRow; i <= endRow; i++) {
                  double x = dv_at(vv, i);
                  if (!isnan(x)) qbuf[qn++] = x;
                }
            } break;
          }
        }
        if (dir[c].end_time >= wEnd) break;
      }
      switch (q->func_id) {
        case FN_SUM:   result = sum; break;
        case FN_RATE_OVER_DELTA:
          /* delta-temporality rate: RateOverDeltaChunkedFunctionD
           * (RateFunctions.scala:424-445) = sum_over_time / window seconds */
          result = sum / (double)(wEnd - wStart) * 1000;
          break;
        case FN_COUNT: result = count; break;
        case FN_AVG:   result = icount > 0 ? sum / icount : (isnan(sum) ? sum : 0); break;
        case FN_MIN:   result = mn; break;
        case FN_MAX:   result = mx; break;
        case FN_STDDEV: case FN_STDVAR: {
          double r;
          if (icount > 0) {
            double avg = sum / icount;
            r = sqsum / icount - avg * avg;
            if (q->func_id == FN_STDDEV) r = sqrt(r);
          } else if (isnan(sum)) r = sum;
          else r = 0;
          result = r;
        } break;
        case FN_CHANGES: result = changes; break;
        case FN_HOLT_WINTERS: result = hwRes; break;
        case FN_LAST: case FN_PRESENT: case FN_TIMESTAMP:
          result = last_val; break;
        case FN_ZSCORE: {
          /* ZScoreChunkedFunctionD (AggrOverTimeFunctions.scala:1592-1603) */
          if (icount > 0) {
            double avg = sum / icount;
            double sd = sqrt(sqsum / icount - avg * avg);
            result = (last_sample - avg) / sd;
          } else if (isnan(sum)) result = sum;
          else result = 0;
        } break;
        case FN_QUANTILE: {
          if (touched && q->param < 0) result = -INFINITY;
          else if (touched && q->param > 1) result = INFINITY;
          else if (qn > 0) {
            qsort(qbuf, (size_t)qn, sizeof(double), cmp_dbl);
            result = interp_quantile(q->param, qbuf, qn);
          }
        } break;
        case FN_PREDICT_LINEAR: {
          /* emit (AggrOverTimeFunctions.scala:1507-1517): counter >= 2 */
          if (plN >= 2) {
            double covXY = plXY - plX * plY / plN;
            double varX = plX2 - plX * plX / plN;
            double slope = covXY / varX;
            double intercept = plY / plN - slope * plX / plN;
            result = slope * q->param + intercept;
          }
        } break;
        case FN_MAD: {
          if (qn > 0) {
            qsort(qbuf, (size_t)qn, sizeof(double), cmp_dbl);
            double median = interp_quantile(0.5, qbuf, qn);
            for (int i = 0; i < qn; i++) qbuf[i] = fabs(median - qbuf[i]);
            qsort(qbuf, (size_t)qn, sizeof(double), cmp_dbl);
            result = interp_quantile(0.5, qbuf, qn);
          }
        } break;
      }
    }
    out[w] = result;
  }
  if (is_rate_family)
    for (int c = 0; c < nchunks; c++)
      if (cr[c].owned) free(cr[c].corrected);
  free(qbuf);
}

/* =========================================================================
 * whole-query execution + fastReduce
 * (AggrOverRangeVectors.scala:320-377; aggregator/{Sum,Count,Min,Max,Avg}RowAggregator)
 * ========================================================================= */
EXPORT int32_t oracle_query_exec(const fdb_view_t* view, const fdb_query_t* q,
                                 double* out, double* out_counts, int32_t nthreads) {
  int nw = (int)((q->end - q->start) / q->step) + 1;
  int ns = view->num_series;
  if (q->agg_id == AGG_NONE) {
#ifdef _OPENMP
    omp_set_num_threads(nthreads > 0 ? nthreads : 1);
    #pragma omp parallel
    {
      eval_ctx_t ctx; ctx.scratch = (double*)malloc(64 * 512 * sizeof(double));
      #pragma omp for schedule(static)
      for (int s = 0; s < ns; s++) eval_series(view, s, q, &ctx, out + (size_t)s * nw);
      free(ctx.scratch);
    }
#else
    eval_ctx_t ctx; ctx.scratch = (double*)malloc(64 * 512 * sizeof(double));
    for (int s = 0; s < ns; s++) eval_series(view, s, q, &ctx, out + (size_t)s * nw);
    free(ctx.scratch);
#endif
    return 0;
  }

  int ng = q->num_groups;
  size_t gridlen = (size_t)ng * nw;

  if (q->agg_id == AGG_TOPK || q->agg_id == AGG_BOTTOMK) {
    /* TopBottomKRowAggregator.scala:29-100: per (group, window) keep the k
     * largest (topk) / smallest (bottomk) non-NaN series values with their
     * keys; NaN padding. Output sorted descending (topk) / ascending. */
    int k = (int)q->param;
    if (k < 1 || k > 16) return -1;
    int top = q->agg_id == AGG_TOPK;
    size_t cells = gridlen * (size_t)k;
    double* ids = out_counts;         /* [G × W × k] series ids as doubles */
    for (size_t i = 0; i < cells; i++) { out[i] = NAN; if (ids) ids[i] = -1; }
    double* row = (double*)malloc((size_t)nw * sizeof(double));
    eval_ctx_t ctx; ctx.scratch = (double*)malloc(64 * 512 * sizeof(double));
    for (int s = 0; s < ns; s++) {
      eval_series(view, s, q, &ctx, row);
      int grp = view->group_ids[s];
      for (int w = 0; w < nw; w++) {
        double x = row[w];
        if (isnan(x)) continue;
        double* cell = out + ((size_t)grp * nw + w) * k;
        double* cid = ids ? ids + ((size_t)grp * nw + w) * k : 0;
        /* insertion into the sorted-k list (k <= 16) */
        int pos = -1;
        for (int j = 0; j < k; j++) {
          if (isnan(cell[j]) || (top ? x > cell[j] : x < cell[j])) { pos = j; break; }
        }
        if (pos >= 0) {
          for (int j = k - 1; j > pos; j--) {
            cell[j] = cell[j - 1];
            if (cid) cid[j] = cid[j - 1];
          }
          cell[pos] = x;
          if (cid) cid[pos] = (double)s;
        }
      }
    }
    free(row); free(ctx.scratch);
    return 0;
  }

  if (q->agg_id == AGG_QUANTILE) {
    /* QuantileRowAggregator (QuantileRowAggregator.scala:21-76): one t-digest
     * per (group, window), samples added in ascending series order, present
     * step emits digest.quantile(q). Partial-mode digest shipping is not
     * implemented (out_counts ignored). */
    if (gridlen > 2000000) return -1;
    tdigest_t* tds = (tdigest_t*)malloc(gridlen * sizeof(tdigest_t));
    if (!tds) return -1;
    for (size_t i = 0; i < gridlen; i++) td_init(&tds[i]);
    double* row = (double*)malloc((size_t)nw * sizeof(double));
    eval_ctx_t ctx; ctx.scratch = (double*)malloc(64 * 512 * sizeof(double));
    for (int s = 0; s < ns; s++) {
      eval_series(view, s, q, &ctx, row);
      int grp = view->group_ids[s];
      for (int w = 0; w < nw; w++)
        td_add(&tds[(size_t)grp * nw + w], row[w]);
    }
    for (size_t i = 0; i < gridlen; i++)
      out[i] = td_quantile(&tds[i], q->param);
    free(tds); free(row); free(ctx.scratch);
    return 0;
  }

  for (size_t i = 0; i < gridlen; i++) out[i] = NAN;
  double* counts = out_counts;
  double* owned_counts = 0;
  if (q->agg_id == AGG_AVG && !counts) {
    owned_counts = (double*)malloc(gridlen * sizeof(double));
    counts = owned_counts;
  }
  if (counts) for (size_t i = 0; i < gridlen; i++) counts[i] = 0;

  int T = nthreads > 0 ? nthreads : 1;
#ifdef _OPENMP
  omp_set_num_threads(T);
#else
  T = 1;
#endif
  /* per-thread partial grids, merged with the same RowAggregator semantics */
  double* pg = (double*)malloc((size_t)T * gridlen * sizeof(double));
  double* pc = (double*)malloc((size_t)T * gridlen * sizeof(double));
  int needs_sq = q->agg_id == AGG_STDDEV || q->agg_id == AGG_STDVAR;
  double* pq = needs_sq ? (double*)malloc((size_t)T * gridlen * sizeof(double)) : 0;
  for (size_t i = 0; i < (size_t)T * gridlen; i++) { pg[i] = NAN; pc[i] = 0; if (pq) pq[i] = 0; }
  double* sq_total = needs_sq ? (double*)calloc(gridlen, sizeof(double)) : 0;

#ifdef _OPENMP
  #pragma omp parallel
#endif
  {
#ifdef _OPENMP
    int t = omp_get_thread_num();
#else
    int t = 0;
#endif
    double* g = pg + (size_t)t * gridlen;
    double* gc = pc + (size_t)t * gridlen;
    double* gq = pq ? pq + (size_t)t * gridlen : 0;
    eval_ctx_t ctx; ctx.scratch = (double*)malloc(64 * 512 * sizeof(double));
    double* row = (double*)malloc((size_t)nw * sizeof(double));
#ifdef _OPENMP
    #pragma omp for schedule(static)
#endif
    for (int s = 0; s < ns; s++) {
      eval_series(view, s, q, &ctx, row);
      int grp = view->group_ids[s];
      double* acc = g + (size_t)grp * nw;
      double* accc = gc + (size_t)grp * nw;
      for (int w = 0; w < nw; w++) {
        double x = row[w];
        if (isnan(x)) continue;   /* NaN rows never contribute (all RowAggregators) */
        accc[w] += 1;
        switch (q->agg_id) {
          case AGG_SUM:   /* SumRowAggregator.scala:23-29 */
            if (isnan(acc[w])) acc[w] = 0;
            acc[w] += x;
            break;
          case AGG_COUNT: case AGG_GROUP:
            /* CountRowAggregator.scala:36-42: sample maps to 1 (0 when NaN); the
             * accumulator leaves NaN only when every mapped value was 0.
             * GROUP folds identically; its present step emits 1
             * (GroupRowAggregator.scala:12-31). */
            if (isnan(acc[w])) acc[w] = 0;
            acc[w] += 1;
            break;
          case AGG_MIN:
            if (isnan(acc[w]) || x < acc[w]) acc[w] = x;
            break;
          case AGG_MAX:
            if (isnan(acc[w]) || x > acc[w]) acc[w] = x;
            break;
          case AGG_AVG:   /* AvgRowAggregator: weighted mean ≡ sum/count */
            if (isnan(acc[w])) acc[w] = 0;
            acc[w] += x;
            break;
          case AGG_STDDEV: case AGG_STDVAR:
            /* StddevRowAggregator.scala:39-58: the running merge is
             * algebraically (sum, sumsq, count) */
            if (isnan(acc[w])) acc[w] = 0;
            acc[w] += x;
            gq[(size_t)grp * nw + w] += x * x;
            break;
        }
      }
    }
    free(ctx.scratch); free(row);
  }

  /* merge thread partials (same op; associative for these aggregators) */
  double* cnt_total = counts ? counts : (double*)calloc(gridlen, sizeof(double));
  for (int t = 0; t < T; t++) {
    double* g = pg + (size_t)t * gridlen;
    double* gc = pc + (size_t)t * gridlen;
    double* gq = pq ? pq + (size_t)t * gridlen : 0;
    for (size_t i = 0; i < gridlen; i++) {
      double x = g[i];
      cnt_total[i] += gc[i];
      if (isnan(x)) continue;
      switch (q->agg_id) {
        case AGG_SUM: case AGG_COUNT: case AGG_AVG:
        case AGG_STDDEV: case AGG_STDVAR: case AGG_GROUP:
          if (isnan(out[i])) out[i] = 0;
          out[i] += x;
          if (sq_total) sq_total[i] += gq[i];
          break;
        case AGG_MIN:
          if (isnan(out[i]) || x < out[i]) out[i] = x;
          break;
        case AGG_MAX:
          if (isnan(out[i]) || x > out[i]) out[i] = x;
          break;
      }
    }
  }
  if (out_counts) {
    /* PARTIAL mode (cross-shard merge inputs, ReduceAggregateExec contract):
     * raw sums with 0 where empty + contribution counts; caller merges then
     * presents (NaN where total count 0; /count for avg). For stddev/stdvar
     * `out` is [2 x G x W]: raw sums then raw sumsq — merging by addition is
     * algebraically StddevRowAggregator.scala:36-52's reduction. */
    if (q->agg_id == AGG_SUM || q->agg_id == AGG_COUNT || q->agg_id == AGG_AVG ||
        q->agg_id == AGG_GROUP || needs_sq)
      for (size_t i = 0; i < gridlen; i++)
        if (isnan(out[i]) && cnt_total[i] == 0) out[i] = 0;
    if (needs_sq)
      for (size_t i = 0; i < gridlen; i++) out[gridlen + i] = sq_total[i];
    /* MIN/MAX partials keep NaN for empty cells; the merging caller maps them
     * to ±inf before the collective. */
  } else if (q->agg_id == AGG_GROUP) {
    /* present: 1 wherever any row contributed (GroupRowAggregator.scala:12-31) */
    for (size_t i = 0; i < gridlen; i++)
      if (!isnan(out[i])) out[i] = 1;
  } else if (q->agg_id == AGG_AVG) {
    /* present: mean = sum/count (AvgRowAggregator.scala:38-46 algebraically) */
    for (size_t i = 0; i < gridlen; i++)
      if (cnt_total[i] > 0) out[i] = out[i] / cnt_total[i];
  } else if (needs_sq) {
    /* present: sqrt(sumsq/n - mean^2) (StddevRowAggregator.scala:49-52) */
    for (size_t i = 0; i < gridlen; i++) {
      if (cnt_total[i] > 0) {
        double mean = out[i] / cnt_total[i];
        double var = sq_total[i] / cnt_total[i] - mean * mean;
        out[i] = q->agg_id == AGG_STDDEV ? sqrt(var) : var;
      }
    }
  }
  free(pg); free(pc);
  if (pq) free(pq);
  if (sq_total) free(sq_total);
  if (!counts) free(cnt_total);
  if (owned_counts) free(owned_counts);
  return 0;
}

/* convenience: evaluate one series only (tests) */
/* CountValuesRowAggregator (CountValuesRowAggregator.scala:26-100): per
 * (group, window) a value→frequency map over non-NaN series results; more
 * than `limit` distinct values is an error (the reference throws at 1000).
 * Output: per cell, n distinct pairs sorted by value ascending. */
EXPORT int32_t oracle_count_values(const fdb_view_t* view, const fdb_query_t* q,
                                   int32_t k_cap, double* out_vals,
                                   double* out_cnts, int32_t* out_n) {
  int nw = (int)((q->end - q->start) / q->step) + 1;
  int ns = view->num_series;
  int ng = q->num_groups;
  size_t gridlen = (size_t)ng * nw;
  if (k_cap < 1 || k_cap > 1000) return -1;
  double* vals = (double*)calloc(gridlen * (size_t)k_cap, sizeof(double));
  double* cnts = (double*)calloc(gridlen * (size_t)k_cap, sizeof(double));
  if (!vals || !cnts) { free(vals); free(cnts); return -1; }
  memset(out_n, 0, gridlen * sizeof(int32_t));
  double* row = (double*)malloc((size_t)nw * sizeof(double));
  eval_ctx_t ctx; ctx.scratch = (double*)malloc(64 * 512 * sizeof(double));
  int rc = 0;
  for (int s = 0; s < ns && rc == 0; s++) {
    eval_series(view, s, q, &ctx, row);
    int grp = view->group_ids[s];
    for (int w = 0; w < nw; w++) {
      double x = row[w];
      if (isnan(x)) continue;
      size_t cell = (size_t)grp * nw + w;
      double* cv = vals + cell * k_cap;
      double* cc = cnts + cell * k_cap;
      int n = out_n[cell];
      /* binary search for x in the sorted distinct list */
      int lo = 0, hi = n;
      while (lo < hi) { int mid = (lo + hi) / 2;
        if (cv[mid] < x) lo = mid + 1; else hi = mid; }
      if (lo < n && cv[lo] == x) { cc[lo] += 1; }
      else {
        if (n >= k_cap) { rc = -2; break; }   /* reference throws at limit */
        for (int j = n; j > lo; j--) { cv[j] = cv[j-1]; cc[j] = cc[j-1]; }
        cv[lo] = x; cc[lo] = 1;
        out_n[cell] = n + 1;
      }
    }
  }
  if (rc == 0) {
    memcpy(out_vals, vals, gridlen * (size_t)k_cap * sizeof(double));
    memcpy(out_cnts, cnts, gridlen * (size_t)k_cap * sizeof(double));
  }
  free(vals); free(cnts); free(row); free(ctx.scratch);
  return rc;
}

EXPORT int32_t oracle_eval_series(const fdb_view_t* view, int32_t sid, const fdb_query_t* q,
                                  double* out) {
  eval_ctx_t ctx; ctx.scratch = (double*)malloc(64 * 512 * sizeof(double));
  eval_series(view, sid, q, &ctx, out);
  free(ctx.scratch);
  return 0;
}

/* decode helpers for unit tests */
EXPORT int32_t oracle_decode_longs(const uint8_t* vec, int64_t* out, int32_t cap) {
  vec_t v; vec_open(vec, &v);
  if (v.n > cap) return -1;
  for (int i = 0; i < v.n; i++) out[i] = lv_at(&v, i);
  return v.n;
}
EXPORT int32_t oracle_decode_doubles(const uint8_t* vec, double* out, int32_t cap) {
  vec_t v; vec_open(vec, &v);
  if (v.n > cap) return -1;
  for (int i = 0; i < v.n; i++) out[i] = dv_at(&v, i);
  return v.n;
}
EXPORT int32_t oracle_vec_info(const uint8_t* vec, int32_t* wf, int32_t* n,
                               int32_t* nbits, int32_t* dropped) {
  vec_t v; vec_open(vec, &v);
  *wf = v.wf; *n = v.n; *nbits = v.nbits; *dropped = v.dropped;
  return 0;
}
EXPORT int32_t oracle_binary_search(const uint8_t* vec, int64_t item) {
  vec_t v; vec_open(vec, &v);
  return lv_binary_search(&v, item);
}
EXPORT double oracle_extrapolated_rate(int64_t ws, int64_t we, int32_t n,
                                       int64_t t1, double v1, int64_t t2, double v2,
                                       int32_t isCounter, int32_t isRate) {
  return extrapolated_rate(ws, we, n, t1, v1, t2, v2, isCounter, isRate);
}

/* =========================================================================
 * sect-delta histogram path (config #4)
 *   vector format  HistogramVector.scala:237-254,491-545; Section.scala:17-24
 *   reader         SectDeltaHistogramReader (HistogramVector.scala:628-737)
 *   rate           HistogramRateFunctionBase (RateFunctions.scala:330-400)
 *   cross-series   HistSumRowAggregator.scala:20-29
 *   quantile       Histogram.quantile (Histogram.scala:63-108)
 * ========================================================================= */
typedef struct {
  int n, nb;
  double first, mult;
  int64_t* cum;      /* [n × nb] raw cumulative bucket values               */
  int64_t* corr;     /* [n × nb] cumulative in-chunk corrections at element */
  int64_t* chunk_corr; /* [nb] total corrections (updateCorrection)         */
} hist_chunk_t;

/* NibblePack delta-decode nb longs from a stream (DeltaSink semantics) */
/* unpackDoubleXOR (NibblePack.scala:360-394): first double raw, the rest
 * XOR-chained through 8-value nibble-packed groups */
EXPORT int32_t oracle_nibblepack_unpack_doubles(const uint8_t* in, int32_t inlen,
                                                double* out, int32_t n) {
  if (n < 1 || inlen < 8) return -1;
  uint64_t last; memcpy(&last, in, 8);
  memcpy(&out[0], &last, 8);
  int pos = 8, i = 1;
  while (i < n) {
    int64_t grp[8]; int consumed;
    if (oracle_nibblepack_unpack8(in + pos, inlen - pos, grp, &consumed) != 0)
      return -1;
    pos += consumed;
    for (int k = 0; k < 8 && i < n; k++, i++) {
      last ^= (uint64_t)grp[k];
      memcpy(&out[i], &last, 8);
    }
  }
  return 0;
}

static int np_unpack_delta(const uint8_t* in, int inlen, int64_t* out, int nb) {
  int pos = 0, i = 0;
  int64_t current = 0;
  while (i < nb) {
    int64_t grp[8]; int consumed;
    if (oracle_nibblepack_unpack8(in + pos, inlen - pos, grp, &consumed) != 0) return -1;
    pos += consumed;
    for (int k = 0; k < 8 && i < nb; k++, i++) { current += grp[k]; out[i] = current; }
  }
  return 0;
}

/* decodes a sect-delta hist vector; returns 0 and fills hc (malloc'd) */
static int hist_open(const uint8_t* p, hist_chunk_t* hc) {
  if (rd_u16(p + 4) != FDB_WF_HIST_SECTDELTA) return -1;
  int n = rd_u16(p + FDB_HIST_OFF_NUMHIST);
  int nb = rd_u16(p + FDB_HIST_OFF_DEF);
  hc->n = n; hc->nb = nb;
  hc->first = rd_f64(p + FDB_HIST_OFF_DEF + 2);
  hc->mult = rd_f64(p + FDB_HIST_OFF_DEF + 10);
  hc->cum = (int64_t*)malloc((size_t)n * nb * 8);
  hc->corr = (int64_t*)calloc((size_t)n * nb, 8);
  hc->chunk_corr = (int64_t*)calloc((size_t)nb, 8);
  const uint8_t* sp = p + FDB_HIST_OFF_DEF + rd_u16(p + FDB_HIST_OFF_DEFSIZE);
  int e = 0;
  int64_t* base = (int64_t*)malloc((size_t)nb * 8);
  int64_t* runc = (int64_t*)calloc((size_t)nb, 8);
  while (e < n) {
    int sbytes = rd_u16(sp);
    int selems = sp[2];
    int stype = sp[3];
    const uint8_t* ep = sp + 4;
    /* TypeDrop section (not first element): correction += value before drop
     * (SectDeltaHistogramReader.corrections, HistogramVector.scala:663-676) */
    if (stype == 1 && e > 0)
      for (int b = 0; b < nb; b++) runc[b] += hc->cum[(size_t)(e - 1) * nb + b];
    for (int se = 0; se < selems && e < n; se++, e++) {
      int elen = rd_u16(ep);
      int64_t* row = hc->cum + (size_t)e * nb;
      if (se == 0) {               /* section base: raw packDelta of cum values */
        if (np_unpack_delta(ep + 2, elen, row, nb) != 0) return -1;
        memcpy(base, row, (size_t)nb * 8);
      } else {                     /* delta element: base + delta-decoded diffs */
        if (np_unpack_delta(ep + 2, elen, row, nb) != 0) return -1;
        for (int b = 0; b < nb; b++) row[b] += base[b];
      }
      memcpy(hc->corr + (size_t)e * nb, runc, (size_t)nb * 8);
      ep += 2 + elen;
    }
    sp += 4 + sbytes;
  }
  memcpy(hc->chunk_corr, runc, (size_t)nb * 8);
  free(base); free(runc);
  return 0;
}
static void hist_close(hist_chunk_t* hc) { free(hc->cum); free(hc->corr); free(hc->chunk_corr); }

/* Histogram.compare (Histogram.scala:204-214), equal schemes: top-down values */
static int hist_less(const int64_t* a, const int64_t* b, int nb) {
  for (int i = nb - 1; i >= 0; i--) {
    if (a[i] != b[i]) return a[i] < b[i];
  }
  return 0;
}

/* Histogram.quantile (Histogram.scala:63-108), geometric buckets */
EXPORT double oracle_hist_quantile(double q, const double* values, int nb,
                                   double first, double mult) {
  if (q < 0) return -INFINITY;
  if (q > 1) return INFINITY;
  double top = nb <= 0 ? NAN : values[nb - 1];
  if (nb < 2 || !(top > 0)) return NAN;
  double rank = q * top;
  int bucket = 0;
  while (values[bucket] < rank) bucket++;
  double bucketStart = bucket == 0 ? 0 : first * pow(mult, bucket - 1);
  double bucketEnd = first * pow(mult, bucket);
  if (bucket == nb - 1 && isinf(bucketEnd)) return first * pow(mult, nb - 2);
  if (bucket == 0 && first <= 0) return first;
  double count = bucket == 0 ? values[0] : values[bucket] - values[bucket - 1];
  rank -= bucket == 0 ? 0 : values[bucket - 1];
  double fraction = rank / count;
  return bucketStart + (bucketEnd - bucketStart) * fraction;
}

/* test helper: decode all cumulative bucket values of a hist vector */
EXPORT int32_t oracle_hist_decode(const uint8_t* vec, int64_t* out, int32_t cap,
                                  int32_t* out_n, int32_t* out_nb) {
  hist_chunk_t hc;
  if (hist_open(vec, &hc) != 0) return -1;
  if (hc.n * hc.nb > cap) { hist_close(&hc); return -1; }
  memcpy(out, hc.cum, (size_t)hc.n * hc.nb * 8);
  *out_n = hc.n; *out_nb = hc.nb;
  hist_close(&hc);
  return 0;
}
EXPORT int32_t oracle_hist_corrections(const uint8_t* vec, int64_t* out, int32_t cap) {
  hist_chunk_t hc;
  if (hist_open(vec, &hc) != 0) return -1;
  if (hc.n * hc.nb > cap) { hist_close(&hc); return -1; }
  memcpy(out, hc.corr, (size_t)hc.n * hc.nb * 8);
  hist_close(&hc);
  return 0;
}

/* histogram_quantile(param, sum by(group)(rate(hist[window]))) — the full
 * config-#4 pipeline. out_bucket_sums [G×W×nb] / out_counts [G×W] /
 * out_quantile [G×W]; any may be NULL. */
/* companion-column variant: adds per-(group,window) max/min over the otel
 * max/min double columns (SumAndMaxOverTimeFuncHD and
 * CumulativeHistRateAndMinMaxFunction, AggrOverTimeFunctions.scala:612-813;
 * cross-series merge maxIgnoreNaN/minIgnoreNaN per HistMaxMinSumAggregator).
 * func_id FN_HIST_RATE -> counter-corrected rate; FN_SUM -> SumOverTime of
 * the histograms (raw bucket sums, no corrections). out_max/out_min NULLable. */
EXPORT int32_t oracle_query_exec_hist_mm(const fdb_view_t* view, const fdb_query_t* q,
                                         int32_t nb, double* sums, double* cnts,
                                         double* out_max, double* out_min,
                                         double* out_quantile);

EXPORT int32_t oracle_query_exec_hist(const fdb_view_t* view, const fdb_query_t* q,
                                      int32_t nb,
                                      double* out_bucket_sums, double* out_counts,
                                      double* out_quantile, int32_t nthreads) {
  int nw = (int)((q->end - q->start) / q->step) + 1;
  int ng = q->num_groups;
  size_t cells = (size_t)ng * nw;
  double* sums = out_bucket_sums ? out_bucket_sums
                                 : (double*)malloc(cells * nb * sizeof(double));
  double* cnts = out_counts ? out_counts : (double*)malloc(cells * sizeof(double));
  for (size_t i = 0; i < cells * nb; i++) sums[i] = 0;
  for (size_t i = 0; i < cells; i++) cnts[i] = 0;

  (void)nthreads;  /* serial: hist oracle runs at test sizes */
  int64_t* lastv = (int64_t*)malloc((size_t)nb * 8);
  int64_t* corr  = (int64_t*)malloc((size_t)nb * 8);
  double* lo = (double*)malloc((size_t)nb * 8);
  double* hi = (double*)malloc((size_t)nb * 8);
  for (int sid = 0; sid < view->num_series; sid++) {
    int first = view->series_first[sid];
    int nchunks = view->series_nchunks[sid];
    if (nchunks > 64) nchunks = 64;
    const fdb_dir_entry_t* dir = view->dir + first;
    vec_t tsv[64];
    hist_chunk_t hv[64];
    for (int c = 0; c < nchunks; c++) {
      vec_open(view->blob + dir[c].ts_off, &tsv[c]);
      if (hist_open(view->blob + dir[c].val_off, &hv[c]) != 0) {
        for (int c2 = 0; c2 < c; c2++) hist_close(&hv[c2]);
        free(lastv); free(corr); free(lo); free(hi);
        return -1;
      }
    }
    int grp = view->group_ids[sid];
    for (int w = 0; w < nw; w++) {
      int64_t wEnd = q->start + (int64_t)w * q->step;
      int64_t wStart = wEnd - q->window;
      int meta_has = 0;
      memset(corr, 0, (size_t)nb * 8);
      int numSamples = 0;
      int64_t lowestTime = INT64_MAX, highestTime = 0;
      int have_lo = 0, have_hi = 0;
      for (int c = 0; c < nchunks; c++) {
        if (dir[c].end_time < wStart) continue;
        const vec_t* tv = &tsv[c];
        const hist_chunk_t* h = &hv[c];
        int startRow = lv_binary_search(tv, wStart) & 0x7fffffff;
        int endRow = lv_ceiling(tv, wEnd);
        if (endRow > dir[c].num_rows - 1) endRow = dir[c].num_rows - 1;
        if (meta_has) {              /* detectDropAndCorrection :654-663 */
          if (hist_less(h->cum, lastv, nb))
            for (int b = 0; b < nb; b++) corr[b] += lastv[b];
        }
        if (startRow <= endRow) {
          int64_t st = lv_at(tv, startRow), en = lv_at(tv, endRow);
          if (st < lowestTime || en > highestTime) {
            numSamples += endRow - startRow + 1;
            if (st < lowestTime) {
              lowestTime = st;
              for (int b = 0; b < nb; b++)
                lo[b] = (double)(h->cum[(size_t)startRow * nb + b]
                                 + h->corr[(size_t)startRow * nb + b] + corr[b]);
              have_lo = 1;
            }
            if (en > highestTime) {
              highestTime = en;
              for (int b = 0; b < nb; b++)
                hi[b] = (double)(h->cum[(size_t)endRow * nb + b]
                                 + h->corr[(size_t)endRow * nb + b] + corr[b]);
              have_hi = 1;
            }
          }
        }
        /* updateCorrection :699-711: lastValue = apply(len-1), the RAW value */
        for (int b = 0; b < nb; b++) {
          corr[b] += h->chunk_corr[b];
          lastv[b] = h->cum[(size_t)(h->n - 1) * nb + b];
        }
        meta_has = 1;
        if (dir[c].end_time >= wEnd) break;
      }
      if (highestTime > lowestTime && have_lo && have_hi) {
        size_t cell = (size_t)grp * nw + w;
        for (int b = 0; b < nb; b++) {
          double r = extrapolated_rate(wStart, wEnd, numSamples,
                                       lowestTime, lo[b], highestTime, hi[b], 1, 1);
          sums[cell * nb + b] += r;   /* HistSumRowAggregator bucket-wise add */
        }
        cnts[cell] += 1;
      }
    }
    for (int c = 0; c < nchunks; c++) hist_close(&hv[c]);
  }
  free(lastv); free(corr); free(lo); free(hi);

  if (out_quantile) {
    double first = 0, mult = 0;
    /* bucket scheme from the first hist chunk */
    for (int sid = 0; sid < view->num_series && first == 0; sid++) {
      const fdb_dir_entry_t* dir = view->dir + view->series_first[sid];
      hist_chunk_t hc;
      if (hist_open(view->blob + dir[0].val_off, &hc) == 0) {
        first = hc.first; mult = hc.mult;
        hist_close(&hc);
      }
    }
    for (size_t i = 0; i < cells; i++) {
      out_quantile[i] = cnts[i] > 0
        ? oracle_hist_quantile(q->param, sums + i * nb, nb, first, mult) : NAN;
    }
  }
  if (!out_bucket_sums) free(sums);
  if (!out_counts) free(cnts);
  return 0;
}


EXPORT int32_t oracle_query_exec_hist_mm(const fdb_view_t* view, const fdb_query_t* q,
                                         int32_t nb, double* sums, double* cnts,
                                         double* out_max, double* out_min,
                                         double* out_quantile) {
  int nw = (int)((q->end - q->start) / q->step) + 1;
  int ng = q->num_groups;
  size_t cells = (size_t)ng * nw;
  int is_rate = q->func_id == 11; /* FN_HIST_RATE */
  for (size_t i = 0; i < cells * nb; i++) sums[i] = 0;
  for (size_t i = 0; i < cells; i++) cnts[i] = 0;
  if (out_max) for (size_t i = 0; i < cells; i++) out_max[i] = NAN;
  if (out_min) for (size_t i = 0; i < cells; i++) out_min[i] = NAN;

  int64_t* lastv = (int64_t*)malloc((size_t)nb * 8);
  int64_t* corr  = (int64_t*)malloc((size_t)nb * 8);
  double* lo = (double*)malloc((size_t)nb * 8);
  double* hi = (double*)malloc((size_t)nb * 8);
  double* wsum = (double*)malloc((size_t)nb * 8);
  for (int sid = 0; sid < view->num_series; sid++) {
    int first = view->series_first[sid];
    int nchunks = view->series_nchunks[sid];
    if (nchunks > 64) nchunks = 64;
    const fdb_dir_entry_t* dir = view->dir + first;
    vec_t tsv[64], mxv[64], mnv[64];
    int have_mm[64];
    hist_chunk_t hv[64];
    for (int c = 0; c < nchunks; c++) {
      vec_open(view->blob + dir[c].ts_off, &tsv[c]);
      have_mm[c] = dir[c].max_off != 0;
      if (have_mm[c]) {
        vec_open(view->blob + dir[c].max_off, &mxv[c]);
        vec_open(view->blob + dir[c].min_off, &mnv[c]);
      }
      if (hist_open(view->blob + dir[c].val_off, &hv[c]) != 0) {
        for (int c2 = 0; c2 < c; c2++) hist_close(&hv[c2]);
        free(lastv); free(corr); free(lo); free(hi); free(wsum);
        return -1;
      }
    }
    int grp = view->group_ids[sid];
    for (int w = 0; w < nw; w++) {
      int64_t wEnd = q->start + (int64_t)w * q->step;
      int64_t wStart = wEnd - q->window;
      int meta_has = 0;
      memset(corr, 0, (size_t)nb * 8);
      int numSamples = 0;
      int64_t lowestTime = INT64_MAX, highestTime = 0;
      int have_lo = 0, have_hi = 0;
      int have_sum = 0;
      double wmax = NAN, wmin = NAN;
      for (int b = 0; b < nb; b++) wsum[b] = 0;
      for (int c = 0; c < nchunks; c++) {
        if (dir[c].end_time < wStart) continue;
        const vec_t* tv = &tsv[c];
        const hist_chunk_t* h = &hv[c];
        int startRow = lv_binary_search(tv, wStart) & 0x7fffffff;
        int endRow = lv_ceiling(tv, wEnd);
        if (endRow > dir[c].num_rows - 1) endRow = dir[c].num_rows - 1;
        if (is_rate && meta_has) {
          if (hist_less(h->cum, lastv, nb))
            for (int b = 0; b < nb; b++) corr[b] += lastv[b];
        }
        if (startRow <= endRow) {
          if (is_rate) {
            int64_t st = lv_at(tv, startRow), en = lv_at(tv, endRow);
            if (st < lowestTime || en > highestTime) {
              numSamples += endRow - startRow + 1;
              if (st < lowestTime) {
                lowestTime = st;
                for (int b = 0; b < nb; b++)
                  lo[b] = (double)(h->cum[(size_t)startRow * nb + b]
                                   + h->corr[(size_t)startRow * nb + b] + corr[b]);
                have_lo = 1;
              }
              if (en > highestTime) {
                highestTime = en;
                for (int b = 0; b < nb; b++)
                  hi[b] = (double)(h->cum[(size_t)endRow * nb + b]
                                   + h->corr[(size_t)endRow * nb + b] + corr[b]);
                have_hi = 1;
              }
            }
          } else {
            /* SumOverTimeChunkedFunctionH: raw histogram add per row */
            for (int i = startRow; i <= endRow; i++)
              for (int b = 0; b < nb; b++)
                wsum[b] += (double)h->cum[(size_t)i * nb + b];
            have_sum = 1;
          }
          if (have_mm[c]) {
            for (int i = startRow; i <= endRow; i++) {
              double mx = dv_at(&mxv[c], i);
              double mn = dv_at(&mnv[c], i);
              if (!isnan(mx) && (isnan(wmax) || mx > wmax)) wmax = mx;
              if (!isnan(mn) && (isnan(wmin) || mn < wmin)) wmin = mn;
            }
          }
        }
        if (is_rate) {
          for (int b = 0; b < nb; b++) {
            corr[b] += h->chunk_corr[b];
            lastv[b] = h->cum[(size_t)(h->n - 1) * nb + b];
          }
          meta_has = 1;
        }
        if (dir[c].end_time >= wEnd) break;
      }
      size_t cell = (size_t)grp * nw + w;
      if (is_rate) {
        if (highestTime > lowestTime && have_lo && have_hi) {
          for (int b = 0; b < nb; b++)
            sums[cell * nb + b] += extrapolated_rate(wStart, wEnd, numSamples,
                                                     lowestTime, lo[b],
                                                     highestTime, hi[b], 1, 1);
          cnts[cell] += 1;
        }
      } else if (have_sum) {
        for (int b = 0; b < nb; b++) sums[cell * nb + b] += wsum[b];
        cnts[cell] += 1;
      }
      /* HistMaxMinSumAggregator: maxIgnoreNaN / minIgnoreNaN merges */
      if (out_max && !isnan(wmax) &&
          (isnan(out_max[cell]) || wmax > out_max[cell])) out_max[cell] = wmax;
      if (out_min && !isnan(wmin) &&
          (isnan(out_min[cell]) || wmin < out_min[cell])) out_min[cell] = wmin;
    }
    for (int c = 0; c < nchunks; c++) hist_close(&hv[c]);
  }
  free(lastv); free(corr); free(lo); free(hi); free(wsum);

  if (out_quantile) {
    double first = 0, mult = 0;
    for (int sid = 0; sid < view->num_series && first == 0; sid++) {
      const fdb_dir_entry_t* dir = view->dir + view->series_first[sid];
      hist_chunk_t hc;
      if (hist_open(view->blob + dir[0].val_off, &hc) == 0) {
        first = hc.first; mult = hc.mult;
        hist_close(&hc);
      }
    }
    for (size_t i = 0; i < cells; i++)
      out_quantile[i] = cnts[i] > 0
        ? oracle_hist_quantile(q->param, sums + i * nb, nb, first, mult) : NAN;
  }
  return 0;
}
