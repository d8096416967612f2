// Per-window sample-buffer functions on the GPU: quantile_over_time,
// median_absolute_deviation_over_time and predict_linear (FN 16/17/18).
//
// These functions need the window's raw sample multiset (sorted, for the
// quantile pair) or a regression over window-relative x values — a different
// kernel shape from the prefix/boundary scans: ONE WAVE PER (series, window)
// pair gathers the window's non-NaN samples from the overlapping chunk
// ranges (the same WindowedChunkIterator chunk rules as everywhere else),
// sorts them in LDS with a bitonic network, and applies the oracle's exact
// epilogue:
//   QuantileOverTimeChunkedFunctionD    AggrOverTimeFunctions.scala:1272-1299
//   MedianAbsoluteDeviationOverTime...  AggrOverTimeFunctions.scala:1302-1330
//   PredictLinearChunkedFunctionD       AggrOverTimeFunctions.scala:1507-1554
// Works on any dataset shape (no chunk summaries needed — search slopes are
// derived from each chunk's first/last timestamps on the fly).
// Capacity: <= WS_MAX_SAMPLES non-NaN samples per window (loud error beyond,
// checked host-side as rows-per-window bound cannot exceed total rows).

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstring>
#include <cmath>

#include "chunk_format.h"
#include "scan_common.h"
#include "../../include/filodb_amd.h"

void fdb_set_error(const char* fmt, ...);   // chunk_builder.cpp

#define WS_MAX_SAMPLES 1024
#define WS_WAVES 4

// first index in [0,n) with ts[i] >= item (lower bound), by interpolation
// guess + exact walk over the encoded vector
__device__ __forceinline__ int ws_search_ge(const DVec& tv, int n, int64_t item,
                                            int64_t ts0, int64_t tsl) {
  if (n <= 0) return 0;
  float inv = (tsl > ts0) ? (float)(n - 1) / (float)(tsl - ts0) : 0.0f;
  int g = (int)((float)(item - ts0) * inv);
  if (g < 0) g = 0;
  if (g > n - 1) g = n - 1;
  while (g > 0 && d_lv_at(&tv, g - 1) >= item) g--;
  while (g < n && d_lv_at(&tv, g) < item) g++;
  return g;
}

template <int FUNC>
__global__ __launch_bounds__(WS_WAVES * 64)
void window_sample_kernel(const uint8_t* __restrict__ blob, DirSoA dir,
                          const int32_t* __restrict__ series_first,
                          const int32_t* __restrict__ series_nchunks,
                          int num_series,
                          int64_t qstart, int64_t qstep, int64_t qwindow,
                          int num_windows, double param, double param2,
                          double* __restrict__ out,
                          int32_t* __restrict__ overflow) {
  __shared__ double buf_all[WS_WAVES][WS_MAX_SAMPLES];
  const int wave = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  double* buf = buf_all[wave];
  const size_t npairs = (size_t)num_series * num_windows;

  for (size_t pair = (size_t)blockIdx.x * WS_WAVES + wave; pair < npairs;
       pair += (size_t)gridDim.x * WS_WAVES) {
    const int sid = (int)(pair / num_windows);
    const int w = (int)(pair % num_windows);
    const int first = series_first[sid];
    const int nchunks = series_nchunks[sid];
    const int64_t wEnd = qstart + (int64_t)w * qstep;
    const int64_t wStart = wEnd - qwindow;

    int qn = 0;
    bool touched = false, over = false;
    double plX = NAN, plY = NAN, plXY = NAN, plX2 = NAN;   // predict_linear
    int plN = 0;
    double hwS = NAN, hwB = NAN, hwNext = NAN, hwRes = NAN; // holt_winters
    const bool collect =
        (FUNC != FN_QUANTILE) || (param >= 0 && param <= 1);

    for (int c = 0; c < nchunks; c++) {
      const int64_t cend = dir.end_time[first + c];
      if (cend < wStart) continue;
      DVec tv, vv;
      int64_t ts0;
      d_vec_open_wide(blob + dir.ts_off[first + c], &tv, &ts0);
      d_vec_open_wide(blob + dir.val_off[first + c], &vv, nullptr);
      const int n = dir.num_rows[first + c];
      const int64_t tsl = dir.end_time[first + c];
      int startRow = (wStart <= ts0) ? 0 : ws_search_ge(tv, n, wStart, ts0, tsl);
      int endRow = (wEnd >= tsl) ? n - 1
                                 : ws_search_ge(tv, n, wEnd + 1, ts0, tsl) - 1;
      if (startRow <= endRow && endRow < n) {
        touched = true;
        if constexpr (FUNC == FN_PREDICT_LINEAR) {
          // regression sums over x=(ts-wEnd)/1000, y=value; lane-parallel
          // accumulation (tree order; 1e-9 tolerance covers the reordering)
          double sx = 0, sy = 0, sxy = 0, sx2 = 0;
          int cnt = 0;
          for (int i = startRow + lane; i <= endRow; i += 64) {
            double y = d_dv_at(&vv, i);
            if (isnan(y)) continue;
            double x = (double)(d_lv_at(&tv, i) - wEnd) / 1000.0;
            sx += x; sy += y; sxy += x * y; sx2 += x * x; cnt++;
          }
          for (int off = 32; off > 0; off >>= 1) {
            sx += __shfl_down(sx, off); sy += __shfl_down(sy, off);
            sxy += __shfl_down(sxy, off); sx2 += __shfl_down(sx2, off);
            cnt += __shfl_down(cnt, off);
          }
          sx = __shfl(sx, 0); sy = __shfl(sy, 0);
          sxy = __shfl(sxy, 0); sx2 = __shfl(sx2, 0); cnt = __shfl(cnt, 0);
          if (cnt > 0) {
            if (isnan(plY)) { plY = sy; plX = sx; plXY = sxy; plX2 = sx2; }
            else { plY += sy; plX += sx; plXY += sxy; plX2 += sx2; }
            plN += cnt;
          }
        } else if constexpr (FUNC == FN_HOLT_WINTERS) {
          // HoltWintersChunkedFunctionD (AggrOverTimeFunctions.scala:
          // 1379-1452), the oracle's exact operation sequence serial on
          // lane 0 (the recurrence has a loop-carried dependency). The
          // reference's one-past-endRow read is modeled as the decoded row
          // when it exists in the chunk and NaN at the chunk's end, exactly
          // like the oracle — bit-identical engine==oracle on any shape;
          // reference parity holds for single-chunk series.
          if (lane == 0) {
            const double sf = param, tf = param2;
            int itPos = startRow, rowNum = startRow;
            if (isnan(hwS) && isnan(hwB)) {
              double s0v = NAN, b0v = NAN;
              int cur = startRow;
              while (cur <= endRow && isnan(s0v)) { s0v = d_dv_at(&vv, itPos++); cur++; }
              while (cur <= endRow && isnan(b0v)) { b0v = d_dv_at(&vv, itPos++); cur++; }
              hwNext = b0v;
              hwB = b0v - s0v;
              hwS = s0v;
              rowNum = cur - 1;
            } else if (isnan(hwB)) {
              double b0v = NAN;
              int cur = startRow;
              while (cur <= endRow && isnan(b0v)) { b0v = d_dv_at(&vv, itPos++); cur++; }
              hwNext = b0v;
              hwB = b0v - hwS;
              rowNum = cur - 1;
            } else {
              itPos++;              // continuation discards one read
            }
            if (!isnan(hwB)) {
              while (rowNum <= endRow) {
                if (!isnan(hwNext)) {
                  double ns = sf * hwNext + (1 - sf) * (hwS + hwB);
                  hwB = tf * (ns - hwS) + (1 - tf) * hwB;
                  hwS = ns;
                }
                hwNext = (itPos < n) ? d_dv_at(&vv, itPos) : NAN;
                itPos++;
                rowNum++;
              }
              hwRes = hwS;
            }
          }
        } else if (collect) {
          // gather the range's non-NaN values into the sort buffer
          for (int base = startRow; base <= endRow; base += 64) {
            const int i = base + lane;
            double x = (i <= endRow) ? d_dv_at(&vv, i) : NAN;
            const bool ok = !isnan(x);
            uint64_t mask = __ballot(ok);
            int here = __popcll(mask);
            if (qn + here > WS_MAX_SAMPLES) { over = true; break; }
            if (ok) {
              int slot = qn + __popcll(mask & ((1ULL << lane) - 1));
              buf[slot] = x;
            }
            qn += here;
          }
          if (over) break;
        }
      }
      if (cend >= wEnd) break;
    }

    double result = NAN;
    if (over) {
      if (lane == 0) atomicExch(overflow, 1);
    } else if constexpr (FUNC == FN_HOLT_WINTERS) {
      result = hwRes;
    } else if constexpr (FUNC == FN_PREDICT_LINEAR) {
      if (plN >= 2) {
        double covXY = plXY - plX * plY / plN;
        double varX = plX2 - plX * plX / plN;
        double slope = covXY / varX;
        double intercept = plY / plN - slope * plX / plN;
        result = slope * param + intercept;
      }
    } else {
      if (FUNC == FN_QUANTILE && touched && param < 0) result = -INFINITY;
      else if (FUNC == FN_QUANTILE && touched && param > 1) result = INFINITY;
      else if (qn > 0) {
        // bitonic sort of buf[0..qn) padded to a power of two with +inf
        d_wait_lds();
        __builtin_amdgcn_wave_barrier();
        int m = 1;
        while (m < qn) m <<= 1;
        for (int i = qn + lane; i < m; i += 64) buf[i] = INFINITY;
        d_wait_lds();
        __builtin_amdgcn_wave_barrier();
        for (int k = 2; k <= m; k <<= 1) {
          for (int j = k >> 1; j > 0; j >>= 1) {
            for (int t = lane; t < m; t += 64) {
              int ixj = t ^ j;
              if (ixj > t) {
                double a = buf[t], b2 = buf[ixj];
                bool up = ((t & k) == 0);
                if ((a > b2) == up) { buf[t] = b2; buf[ixj] = a; }
              }
            }
            d_wait_lds();
            __builtin_amdgcn_wave_barrier();
          }
        }
        // interp_quantile (oracle-exact): rank = q*(n-1)
        auto interp = [&](double q, int n2) {
          double rank = q * (n2 - 1);
          int lower = (int)floor(rank);
          if (lower < 0) lower = 0;
          int upper = lower + 1 < n2 - 1 ? lower + 1 : n2 - 1;
          double weight = rank - floor(rank);
          return buf[lower] * (1 - weight) + buf[upper] * weight;
        };
        if constexpr (FUNC == FN_QUANTILE) {
          result = interp(param, qn);
        } else {  // FN_MAD: |median - x| re-sorted, median again
          double median = interp(0.5, qn);
          d_wait_lds();
          __builtin_amdgcn_wave_barrier();
          for (int i = lane; i < qn; i += 64) buf[i] = fabs(median - buf[i]);
          for (int i = qn + lane; i < m; i += 64) buf[i] = INFINITY;
          d_wait_lds();
          __builtin_amdgcn_wave_barrier();
          for (int k = 2; k <= m; k <<= 1) {
            for (int j = k >> 1; j > 0; j >>= 1) {
              for (int t = lane; t < m; t += 64) {
                int ixj = t ^ j;
                if (ixj > t) {
                  double a = buf[t], b2 = buf[ixj];
                  bool up = ((t & k) == 0);
                  if ((a > b2) == up) { buf[t] = b2; buf[ixj] = a; }
                }
              }
              d_wait_lds();
              __builtin_amdgcn_wave_barrier();
            }
          }
          result = interp(0.5, qn);
        }
      }
    }
    if (lane == 0) out[pair] = result;
    d_wait_lds();
    __builtin_amdgcn_wave_barrier();
  }
}

bool fdb_window_sample_supported(int func_id) {
  return func_id == FN_QUANTILE || func_id == FN_MAD ||
         func_id == FN_PREDICT_LINEAR || func_id == FN_HOLT_WINTERS;
}

int32_t fdb_launch_window_sample(hipStream_t stream, const uint8_t* blob,
                                 DirSoA dir, const int32_t* series_first,
                                 const int32_t* series_nchunks, int num_series,
                                 int64_t qstart, int64_t qstep, int64_t qwindow,
                                 int num_windows, int func_id, double param,
                                 double param2, double* out, int32_t* overflow) {
  size_t pairs = (size_t)num_series * num_windows;
  size_t grid = (pairs + WS_WAVES - 1) / WS_WAVES;
  if (grid > 8192) grid = 8192;
  if (grid < 1) grid = 1;
  #define WARGS blob, dir, series_first, series_nchunks, num_series, \
      qstart, qstep, qwindow, num_windows, param, param2, out, overflow
  switch (func_id) {
    case FN_QUANTILE:
      hipLaunchKernelGGL((window_sample_kernel<FN_QUANTILE>), dim3((uint32_t)grid),
                         dim3(WS_WAVES * 64), 0, stream, WARGS);
      break;
    case FN_MAD:
      hipLaunchKernelGGL((window_sample_kernel<FN_MAD>), dim3((uint32_t)grid),
                         dim3(WS_WAVES * 64), 0, stream, WARGS);
      break;
    case FN_PREDICT_LINEAR:
      hipLaunchKernelGGL((window_sample_kernel<FN_PREDICT_LINEAR>),
                         dim3((uint32_t)grid), dim3(WS_WAVES * 64), 0, stream,
                         WARGS);
      break;
    case FN_HOLT_WINTERS:
      hipLaunchKernelGGL((window_sample_kernel<FN_HOLT_WINTERS>),
                         dim3((uint32_t)grid), dim3(WS_WAVES * 64), 0, stream,
                         WARGS);
      break;
    default:
      fdb_set_error("window sample: unsupported func_id %d", func_id);
      return FDB_ERR_BADARG;
  }
  #undef WARGS
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    fdb_set_error("window_sample_kernel launch failed: %s", hipGetErrorString(e));
    return FDB_ERR;
  }
  return FDB_OK;
}
