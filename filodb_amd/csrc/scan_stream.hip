// Streaming general path: multi-chunk series with NO capacity caps.
//
// Round 1 kept whole series LDS-resident (two tiers, capped at 1600 rows /
// 16 chunks — a 24h@15s lookback could not run). Round 2 replaces both tiers:
//
//  * summary_kernel (once per dataset upload, one wave per chunk): decodes
//    each chunk into LDS and reduces it to a 192-B ChunkSum — totals
//    (sum/sumsq/count/min/max/interior-changes), boundary values, the
//    counter-correction pieces (in-chunk correction total, sparse reset
//    table, updateCorrection operand), and the interpolation-search slope.
//    Query-independent, amortized across queries.
//  * stream_walk_kernel (per query): one wave per series, lanes split the
//    windows; each window walks its overlapping chunks exactly like the
//    reference's WindowedChunkIterator + addChunks chain
//    (ChunkSetInfo.scala:445-529, RangeFunction.scala:131-198), taking FULL
//    middle chunks from their summaries in O(1) and decoding only the two
//    boundary ranges directly from the packed vectors in global memory
//    (L1/L2-served — the walk revisits the same small chunks across windows).
//
// Per-window semantics are the round-1 general path's (parity-green against
// the oracle): the NaN-poison sum quirk, the single-row-NaN chunk skip for
// counters, the correction meta chain, LastSample/present/timestamp rules.
// Unbounded: rows per series, chunks per series, window/step ratio and
// num_windows. Per-chunk rows stay <= 400 (the reference's own chunk cap,
// filodb-defaults.conf:835).

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstring>
#include <cmath>

#include "chunk_format.h"
#include "scan_common.h"
#include "../../include/filodb_amd.h"

void fdb_set_error(const char* fmt, ...);   // chunk_builder.cpp

#define SUM_MAX_ROWS 400
#define SUM_DROPS 8          // sparse per-chunk counter-reset table capacity

struct ChunkSum {            // 192 B per chunk, in a device-resident array
  int64_t ts0, ts_last;      // decoded first/last timestamps
  double sum, sqsum;         // NaN-zeroed totals
  double first_val, last_val;   // RAW val[0], val[n-1]
  double vmin, vmax;         // NaN-ignoring (NaN when all-NaN)
  double chunk_corr;         // CorrectingDoubleVectorReader total correction
  double last_for_update;    // updateCorrection operand (DoubleVector.scala:375-391)
  double dcum[SUM_DROPS];    // cumulative correction at/after dpos[j]
  int16_t dpos[SUM_DROPS];   // reset positions (ascending)
  int32_t cnt;               // non-NaN count
  int32_t changes_in;        // interior value changes (NaN-aware)
  float   inv_slope;         // (n-1)/(ts_last-ts0); 0 when degenerate
  uint8_t v0_nan, dropped, dense, ndrops;
  int32_t _pad[3];
};
static_assert(sizeof(ChunkSum) == 192, "ChunkSum layout");

// ---------------------------------------------------------------------------
// summary kernel: one wave per chunk
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256)
void summary_kernel(const uint8_t* __restrict__ blob, DirSoA dir,
                    int64_t num_chunks, ChunkSum* __restrict__ out) {
  __shared__ int64_t ts_all[4][SUM_MAX_ROWS];
  __shared__ double val_all[4][SUM_MAX_ROWS];
  __shared__ int16_t dpos_all[4][SUM_DROPS];
  __shared__ double dcum_all[4][SUM_DROPS];
  const int wave = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  int64_t* ts = ts_all[wave];
  double* val = val_all[wave];

  for (int64_t c = blockIdx.x * 4 + wave; c < num_chunks; c += gridDim.x * 4) {
    DVec tv, vv;
    d_vec_open_wide(blob + dir.ts_off[c], &tv, nullptr);
    d_vec_open_wide(blob + dir.val_off[c], &vv, nullptr);
    int n = dir.num_rows[c];
    if (n > SUM_MAX_ROWS || n > tv.n) n = 0;
    if (n > 0) {
      d_decode_chunk<false>(tv, n, ts, nullptr, lane);
      d_decode_chunk<true>(vv, n, nullptr, val, lane);
    }
    d_wait_lds();
    __builtin_amdgcn_wave_barrier();

    ChunkSum s;
    memset(&s, 0, sizeof(s));
    s.vmin = NAN; s.vmax = NAN; s.first_val = NAN; s.last_val = NAN;
    if (n > 0) {
      double sum = 0, sq = 0, mn = NAN, mx = NAN;
      int cnt = 0, changes = 0;
      double carry_raw = NAN;                 // previous row's raw value
      double corr_carry = 0;                  // drop-scan carries
      double corr_x = -1.7976931348623157e308;
      int ndrops = 0;
      bool dense = false;
      const bool dropped = vv.dropped != 0;
      double last_nonnan = 0;                 // last_for_update when dropped
      int last_nonnan_idx = -1;
      for (int base = 0; base < n; base += 64) {
        const int i = base + lane;
        const bool live = i < n;
        double raw = live ? val[i] : NAN;
        bool ok = live && !isnan(raw);
        double x = ok ? raw : 0;
        sum += x;                              // lane partials; reduced below
        sq += x * x;
        cnt += ok ? 1 : 0;
        if (ok && (isnan(mn) || raw < mn)) mn = raw;
        if (ok && (isnan(mx) || raw > mx)) mx = raw;
        // interior changes (DoubleVectorDataReader64.changes :283-302)
        double px = __shfl_up(raw, 1);
        if (lane == 0) px = carry_raw;
        changes += (i > 0 && live && !isnan(raw) && !isnan(px) && raw != px)
                       ? 1 : 0;
        carry_raw = __shfl(raw, 63);
        // counter-reset scan (NaN→0; CorrectingDoubleVectorReader :325-342);
        // table slots published through LDS so lane 0 can emit them
        if (dropped) {
          double pz = __shfl_up(x, 1);
          if (lane == 0) pz = corr_x;
          double ci = (live && x < pz) ? pz : 0;
          double scan = wave_incl_scan(ci, lane);
          uint64_t mask = __ballot(ci != 0);
          int here = __popcll(mask);
          if (here) {
            if (ndrops + here > SUM_DROPS) {
              dense = true;
            } else if (ci != 0) {
              int slot = ndrops + __popcll(mask & ((1ULL << lane) - 1));
              dpos_all[wave][slot] = (int16_t)i;
              dcum_all[wave][slot] = corr_carry + scan;
            }
            if (!dense) ndrops += here;
          }
          corr_carry += __shfl(scan, 63);
          corr_x = __shfl(x, 63);
          if (ok) { last_nonnan_idx = i; last_nonnan = raw; }
        }
      }
      for (int off = 32; off > 0; off >>= 1) {
        sum += __shfl_down(sum, off);
        sq += __shfl_down(sq, off);
        cnt += __shfl_down(cnt, off);
        changes += __shfl_down(changes, off);
        double o = __shfl_down(mn, off);
        if (!isnan(o) && (isnan(mn) || o < mn)) mn = o;
        o = __shfl_down(mx, off);
        if (!isnan(o) && (isnan(mx) || o > mx)) mx = o;
        int oi = __shfl_down(last_nonnan_idx, off);
        double ov = __shfl_down(last_nonnan, off);
        if (oi > last_nonnan_idx) { last_nonnan_idx = oi; last_nonnan = ov; }
      }
      d_wait_lds();
      __builtin_amdgcn_wave_barrier();
      s.ts0 = ts[0];
      s.ts_last = ts[n - 1];
      s.sum = sum; s.sqsum = sq; s.cnt = cnt; s.changes_in = changes;
      s.vmin = mn; s.vmax = mx;
      s.first_val = val[0];
      s.last_val = val[n - 1];
      s.v0_nan = isnan(val[0]) ? 1 : 0;
      s.dropped = dropped ? 1 : 0;
      s.dense = dense ? 1 : 0;
      s.ndrops = (uint8_t)(dense ? 0 : ndrops);
      s.chunk_corr = corr_carry;
      s.last_for_update = dropped ? (last_nonnan_idx >= 0 ? last_nonnan : 0)
                                  : val[n - 1];
      s.inv_slope = (s.ts_last > s.ts0)
                        ? (float)(n - 1) / (float)(s.ts_last - s.ts0) : 0.0f;
      if (!dense) {
        for (int j = 0; j < ndrops; j++) {
          s.dpos[j] = dpos_all[wave][j];
          s.dcum[j] = dcum_all[wave][j];
        }
      }
    }
    if (lane == 0) out[c] = s;
    d_wait_lds();
    __builtin_amdgcn_wave_barrier();
  }
}

// ---------------------------------------------------------------------------
// global-memory row access for the walk kernel's boundary ranges
// ---------------------------------------------------------------------------
// first index in [0,n) with ts[i] >= item; n when none — interpolation guess
// + exact walk over the encoded vector (DeltaDeltaDataReader.binarySearch
// semantics for our lower-bound convention)
__device__ __forceinline__ int g_search_ge(const DVec& tv, int n, int64_t item,
                                           int64_t ts0, float inv_slope) {
  if (n <= 0) return 0;
  int g = (int)((float)(item - ts0) * inv_slope);
  if (g < 0) g = 0;
  if (g > n - 1) g = n - 1;
  while (g > 0 && d_lv_at(&tv, g - 1) >= item) g--;
  while (g < n && d_lv_at(&tv, g) < item) g++;
  return g;
}

// in-chunk corrected value at row (CorrectingDoubleVectorReader :305-392):
// raw NaN→0 plus the correction step function from the sparse table, or a
// serial recompute when the chunk overflowed the table
__device__ double g_corrected(const DVec& vv, const ChunkSum& cs, int row) {
  double x = d_dv_at(&vv, row);
  if (!cs.dropped) return x;
  if (isnan(x)) x = 0;
  if (cs.dense) {
    double corr = 0, last = -1.7976931348623157e308;
    for (int j = 0; j <= row; j++) {
      double v = d_dv_at(&vv, j);
      if (isnan(v)) v = 0;
      if (v < last) corr += last;
      last = v;
    }
    return x + corr;
  }
  double corr = 0;
  for (int j = 0; j < cs.ndrops; j++) {
    if (cs.dpos[j] <= row) corr = cs.dcum[j]; else break;
  }
  return x + corr;
}

// ---------------------------------------------------------------------------
// the walk kernel: one wave per series, lanes split windows; per (window,
// chunk): summaries for full chunks, direct decode for boundary ranges
// ---------------------------------------------------------------------------
template <int FUNC>
__global__ __launch_bounds__(256)
void stream_walk_kernel(const uint8_t* __restrict__ blob, DirSoA dir,
                        const ChunkSum* __restrict__ sums,
                        const int32_t* __restrict__ series_first,
                        const int32_t* __restrict__ series_nchunks,
                        int num_series,
                        int64_t qstart, int64_t qstep, int64_t qend,
                        int64_t qwindow, int num_windows,
                        double* __restrict__ out) {
  constexpr bool RATE_FAMILY = (FUNC <= FN_DELTA);
  const int wave = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;

  for (int sid = blockIdx.x * 4 + wave; sid < num_series;
       sid += gridDim.x * 4) {
    const int first = series_first[sid];
    const int nchunks = series_nchunks[sid];

    for (int w = lane; w < num_windows; w += 64) {
      const int64_t wEnd = qstart + (int64_t)w * qstep;
      const int64_t wStart = wEnd - qwindow;
      double result = NAN;

      if constexpr (RATE_FAMILY) {
        bool meta_has = false;
        double meta_last = 0, meta_corr = 0;
        int numSamples = 0;
        int64_t lowestTime = INT64_MAX, highestTime = 0;
        double lowestValue = NAN, highestValue = NAN;
        constexpr bool isCounter = (FUNC != FN_DELTA);
        for (int c = 0; c < nchunks; c++) {
          const int64_t cend = dir.end_time[first + c];
          if (cend < wStart) continue;        // WindowedChunkIterator drop rule
          const ChunkSum& cs = sums[first + c];
          const int n = dir.num_rows[first + c];
          DVec tv, vv;
          d_vec_open_wide(blob + dir.ts_off[first + c], &tv, nullptr);
          d_vec_open_wide(blob + dir.val_off[first + c], &vv, nullptr);
          int startRow = (wStart <= cs.ts0) ? 0
              : g_search_ge(tv, n, wStart, cs.ts0, cs.inv_slope);
          int endRow = (wEnd >= cs.ts_last) ? n - 1
              : g_search_ge(tv, n, wEnd + 1, cs.ts0, cs.inv_slope) - 1;
          if (isCounter && meta_has) {        // detectDropAndCorrection
            double firstv = cs.first_val;
            if (isnan(firstv) || firstv < meta_last) meta_corr += meta_last;
          }
          if (startRow <= endRow && endRow < n) {
            bool skip = isCounter && startRow == 0 && endRow == 0 && cs.v0_nan;
            if (!skip) {
              int64_t st = (startRow == 0) ? cs.ts0 : d_lv_at(&tv, startRow);
              int64_t en = (endRow == n - 1) ? cs.ts_last : d_lv_at(&tv, endRow);
              if (st < lowestTime || en > highestTime) {
                numSamples += endRow - startRow + 1;
                if (st < lowestTime) {
                  lowestTime = st;
                  lowestValue = isCounter
                      ? g_corrected(vv, cs, startRow) + meta_corr
                      : d_dv_at(&vv, startRow);
                }
                if (en > highestTime) {
                  highestTime = en;
                  highestValue = isCounter
                      ? g_corrected(vv, cs, endRow) + meta_corr
                      : d_dv_at(&vv, endRow);
                }
              }
            }
          }
          if (isCounter) {                    // updateCorrection
            if (cs.dropped) meta_corr += cs.chunk_corr;
            meta_last = cs.last_for_update;
            meta_has = true;
          }
          if (cend >= wEnd) break;            // add-while rule
        }
        if (highestTime > lowestTime)
          result = d_extrapolated_rate(wStart, wEnd, numSamples,
                                       lowestTime, lowestValue,
                                       highestTime, highestValue,
                                       isCounter, FUNC == FN_RATE);
      } else {
        double sum = NAN, sqsum = NAN, mm = NAN;
        double changes = NAN, prev = NAN;
        double last_val = NAN, last_sample = NAN;
        int64_t last_ts = -1;
        int icount = 0;
        for (int c = 0; c < nchunks; c++) {
          const int64_t cend = dir.end_time[first + c];
          if (cend < wStart) continue;
          const ChunkSum& cs = sums[first + c];
          const int n = dir.num_rows[first + c];
          DVec tv, vv;
          d_vec_open_wide(blob + dir.ts_off[first + c], &tv, nullptr);
          d_vec_open_wide(blob + dir.val_off[first + c], &vv, nullptr);
          int startRow = (wStart <= cs.ts0) ? 0
              : g_search_ge(tv, n, wStart, cs.ts0, cs.inv_slope);
          int endRow = (wEnd >= cs.ts_last) ? n - 1
              : g_search_ge(tv, n, wEnd + 1, cs.ts0, cs.inv_slope) - 1;

          if constexpr (FUNC == FN_LAST || FUNC == FN_PRESENT) {
            // LastSampleChunkedFunction.addChunks (RangeFunction.scala:599-614)
            if (endRow >= 0 && endRow < n) {
              int64_t t = (endRow == n - 1) ? cs.ts_last : d_lv_at(&tv, endRow);
              if (t >= wStart && t > last_ts) {
                double v = d_dv_at(&vv, endRow);
                if (FUNC == FN_LAST) { last_ts = t; last_val = v; }
                else if (!isnan(v)) { last_ts = t; last_val = 1.0; }
                else if (endRow > 0) {
                  last_ts = t;
                  last_val = isnan(d_dv_at(&vv, endRow - 1)) ? NAN : 1.0;
                }
              }
            }
            if (cend >= wEnd) break;
            continue;
          }
          if constexpr (FUNC == FN_TIMESTAMP) {
            if (endRow >= 0 && endRow < n) {
              int64_t t = (endRow == n - 1) ? cs.ts_last : d_lv_at(&tv, endRow);
              if (t > last_ts) { last_ts = t; last_val = (double)t / 1000.0; }
            }
            if (cend >= wEnd) break;
            continue;
          }

          if (startRow <= endRow && endRow < n) {
            const bool full = (startRow == 0 && endRow == n - 1);
            if constexpr (FUNC == FN_SUM || FUNC == FN_AVG ||
                          FUNC == FN_RATE_OVER_DELTA || FUNC == FN_COUNT) {
              double csum; int cc;
              if (full) { csum = cs.cnt > 0 ? cs.sum : NAN; cc = cs.cnt; }
              else {
                csum = NAN; cc = 0;
                for (int i = startRow; i <= endRow; i++) {
                  double x = d_dv_at(&vv, i);
                  if (isnan(x)) continue;
                  if (isnan(csum)) csum = 0;
                  csum += x; cc++;
                }
              }
              if constexpr (FUNC == FN_COUNT) {
                if (isnan(sum)) sum = 0;     // CountOverTime: any range starts 0
                sum += (double)cc;
              } else {
                if (!isnan(csum) && isnan(sum)) sum = 0;
                sum += csum;                  // NaN-poison quirk preserved
                icount += cc;
              }
            } else if constexpr (FUNC == FN_MIN || FUNC == FN_MAX) {
              constexpr bool IS_MIN = (FUNC == FN_MIN);
              auto acc = [&](double x) {
                if (!isnan(x) && (isnan(mm) || (IS_MIN ? x < mm : x > mm)))
                  mm = x;
              };
              if (full) acc(IS_MIN ? cs.vmin : cs.vmax);
              else for (int i = startRow; i <= endRow; i++) acc(d_dv_at(&vv, i));
            } else if constexpr (FUNC == FN_STDDEV || FUNC == FN_STDVAR ||
                                 FUNC == FN_ZSCORE) {
              // VarOverTimeChunkedFunctionD :1082-1115; last_sample set only
              // from a non-NaN range end (:1103)
              double csm = NAN, csq = NAN; int cc = 0;
              double end_raw;
              if (full) {
                if (cs.cnt > 0) { csm = cs.sum; csq = cs.sqsum; cc = cs.cnt; }
                end_raw = cs.last_val;
              } else {
                for (int i = startRow; i <= endRow; i++) {
                  double x = d_dv_at(&vv, i);
                  if (isnan(x)) continue;
                  if (isnan(csm)) { csm = 0; csq = 0; }
                  csm += x; csq += x * x; cc++;
                }
                end_raw = d_dv_at(&vv, endRow);
              }
              if (!isnan(end_raw)) last_sample = end_raw;
              if (!isnan(csm) && isnan(sum)) sum = 0;
              sum += csm;
              if (!isnan(csq) && isnan(sqsum)) sqsum = 0;
              sqsum += csq;
              icount += cc;
            } else {  // FN_CHANGES
              if (isnan(changes)) changes = 0;
              double first_raw, last_raw;
              double ch;
              if (full) {
                ch = (double)cs.changes_in;
                first_raw = cs.first_val; last_raw = cs.last_val;
              } else {
                ch = 0;
                double p = NAN;
                for (int i = startRow; i <= endRow; i++) {
                  double x = d_dv_at(&vv, i);
                  if (i > startRow && !isnan(x) && !isnan(p) && x != p) ch += 1;
                  p = x;
                }
                first_raw = d_dv_at(&vv, startRow);
                last_raw = p;
              }
              if (!isnan(first_raw) && !isnan(prev) && first_raw != prev)
                ch += 1;                      // cross-chunk boundary pair
              changes += ch;
              prev = last_raw;
            }
          }
          if (cend >= wEnd) break;
        }
        switch (FUNC) {
          case FN_SUM: result = sum; break;
          case FN_RATE_OVER_DELTA:
            // RateOverDeltaChunkedFunctionD (RateFunctions.scala:424-445)
            result = sum / (double)(wEnd - wStart) * 1000;
            break;
          case FN_COUNT: result = sum; break;
          case FN_AVG:
            result = icount > 0 ? sum / icount : (isnan(sum) ? sum : 0);
            break;
          case FN_MIN: case FN_MAX: result = mm; break;
          case FN_STDDEV: case FN_STDVAR: {
            if (icount > 0) {
              double avg = sum / icount;
              double r = sqsum / icount - avg * avg;
              result = (FUNC == FN_STDDEV) ? sqrt(r) : r;
            } else result = isnan(sum) ? sum : 0;
          } break;
          case FN_CHANGES: result = changes; break;
          case FN_LAST: case FN_PRESENT: case FN_TIMESTAMP:
            result = last_val; break;
          case FN_ZSCORE: {
            if (icount > 0) {
              double avg = sum / icount;
              double sd = sqrt(sqsum / icount - avg * avg);
              result = (last_sample - avg) / sd;
            } else result = isnan(sum) ? sum : 0;
          } break;
        }
      }
      out[(size_t)sid * num_windows + w] = result;
    }
  }
}

// ---------------------------------------------------------------------------
// host-side launchers
// ---------------------------------------------------------------------------
int32_t fdb_launch_summaries(hipStream_t stream, const uint8_t* blob, DirSoA dir,
                             int64_t num_chunks, void* sums) {
  int grid = (int)((num_chunks + 3) / 4);
  if (grid > 8192) grid = 8192;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(summary_kernel, dim3(grid), dim3(256), 0, stream,
                     blob, dir, num_chunks, (ChunkSum*)sums);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    fdb_set_error("summary_kernel launch failed: %s", hipGetErrorString(e));
    return FDB_ERR;
  }
  return FDB_OK;
}

int64_t fdb_chunksum_bytes(int64_t num_chunks) {
  return num_chunks * (int64_t)sizeof(ChunkSum);
}

bool fdb_stream_walk_supported(int func_id) {
  switch (func_id) {
    case FN_RATE: case FN_INCREASE: case FN_DELTA:
    case FN_SUM: case FN_COUNT: case FN_AVG: case FN_RATE_OVER_DELTA:
    case FN_MIN: case FN_MAX: case FN_STDDEV: case FN_STDVAR:
    case FN_CHANGES: case FN_LAST: case FN_PRESENT: case FN_TIMESTAMP:
    case FN_ZSCORE:
      return true;
    default:
      return false;
  }
}

int32_t fdb_launch_stream_walk(hipStream_t stream, const uint8_t* blob,
                               DirSoA dir, const void* sums,
                               const int32_t* series_first,
                               const int32_t* series_nchunks, int num_series,
                               int64_t qstart, int64_t qstep, int64_t qend,
                               int64_t qwindow, int num_windows, int func_id,
                               double* out) {
  int grid = (num_series + 3) / 4;
  if (grid > 8192) grid = 8192;
  if (grid < 1) grid = 1;
  #define SARGS blob, dir, (const ChunkSum*)sums, series_first, series_nchunks, \
      num_series, qstart, qstep, qend, qwindow, num_windows, out
  #define SCASE(F) case F: \
    hipLaunchKernelGGL((stream_walk_kernel<F>), dim3(grid), dim3(256), 0, \
                       stream, SARGS); break
  switch (func_id) {
    SCASE(FN_RATE); SCASE(FN_INCREASE); SCASE(FN_DELTA); SCASE(FN_SUM);
    SCASE(FN_COUNT); SCASE(FN_AVG); SCASE(FN_RATE_OVER_DELTA);
    SCASE(FN_MIN); SCASE(FN_MAX); SCASE(FN_STDDEV); SCASE(FN_STDVAR);
    SCASE(FN_CHANGES); SCASE(FN_LAST); SCASE(FN_PRESENT); SCASE(FN_TIMESTAMP);
    SCASE(FN_ZSCORE);
    default:
      fdb_set_error("stream walk: unsupported func_id %d", func_id);
      return FDB_ERR_BADARG;
  }
  #undef SCASE
  #undef SARGS
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    fdb_set_error("stream_walk_kernel launch failed: %s", hipGetErrorString(e));
    return FDB_ERR;
  }
  return FDB_OK;
}
