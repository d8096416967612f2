// Shared device-side pieces of the MI355X scan engine: frozen-vector readers,
// wave scans, the extrapolatedRate epilogue and launch plumbing types.
// Layouts: DESIGN.md §2 (bit-faithful to the reference's off-heap vectors).
#ifndef FDB_SCAN_COMMON_H
#define FDB_SCAN_COMMON_H

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstring>
#include <cmath>

#include "chunk_format.h"

// function / aggregation ids (include/filodb_amd.h; dispatch table
// RangeFunction.scala:294-410, aggregator/RowAggregator.scala)
enum { FN_RATE=0, FN_INCREASE=1, FN_DELTA=2, FN_SUM=3, FN_COUNT=4, FN_AVG=5,
       FN_MIN=6, FN_MAX=7, FN_STDDEV=8, FN_STDVAR=9, FN_CHANGES=10, FN_LAST=12,
       FN_PRESENT=13, FN_TIMESTAMP=14, FN_ZSCORE=15,
       FN_QUANTILE=16, FN_MAD=17, FN_PREDICT_LINEAR=18, FN_RATE_OVER_DELTA=19,
       FN_HOLT_WINTERS=20 };
enum { AGG_NONE=0, AGG_SUM=1, AGG_COUNT=2, AGG_MIN=3, AGG_MAX=4, AGG_AVG=5,
       AGG_TOPK=6, AGG_BOTTOMK=7, AGG_STDDEV=8, AGG_STDVAR=9, AGG_GROUP=10,
       AGG_QUANTILE=11, AGG_COUNT_VALUES=12 };

struct DirSoA {
  const uint64_t* ts_off;
  const uint64_t* val_off;
  const int64_t*  start_time;
  const int64_t*  end_time;
  const int32_t*  num_rows;
};

__device__ __forceinline__ uint16_t d_u16(const uint8_t* p) { uint16_t v; memcpy(&v, p, 2); return v; }
__device__ __forceinline__ uint32_t d_u32(const uint8_t* p) { uint32_t v; memcpy(&v, p, 4); return v; }
__device__ __forceinline__ int32_t  d_i32(const uint8_t* p) { int32_t v; memcpy(&v, p, 4); return v; }
__device__ __forceinline__ int64_t  d_i64(const uint8_t* p) { int64_t v; memcpy(&v, p, 8); return v; }
__device__ __forceinline__ double   d_f64(const uint8_t* p) { double v; memcpy(&v, p, 8); return v; }

struct DVec {           // opened vector header
  const uint8_t* idata;
  int64_t init;
  int32_t slope;
  int n;
  uint16_t wf;
  uint8_t nbits, sign, dropped;
};

__device__ inline void d_vec_open(const uint8_t* p, DVec* v) {
  v->wf = d_u16(p + 4);
  v->dropped = (d_u16(p + 6) & FDB_DROP_MASK) != 0;
  if (v->wf == FDB_WF_DDV) {
    v->init = d_i64(p + FDB_DDV_OFF_INIT);
    v->slope = d_i32(p + FDB_DDV_OFF_SLOPE);
    const uint8_t* inner = p + FDB_DDV_OFF_INNER;
    v->nbits = inner[6] & FDB_NBITS_MASK;
    v->sign = (inner[6] & FDB_SIGN_MASK) != 0;
    v->idata = inner + FDB_PRIM_OFF_DATA;
    int numBytes = (int)d_u32(inner);
    int bitShift = inner[7] & 0x3f;
    v->n = ((numBytes - 4) * 8 + (bitShift != 0 ? bitShift - 8 : 0)) / v->nbits;
  } else if (v->wf == FDB_WF_DDV_CONST) {
    v->n = d_i32(p + FDB_DDVC_OFF_NELEM);
    v->init = d_i64(p + FDB_DDVC_OFF_INIT);
    v->slope = d_i32(p + FDB_DDVC_OFF_SLOPE);
    v->idata = nullptr; v->nbits = 0; v->sign = 0;
  } else {
    v->n = ((int)d_u32(p) - 4) / 8;
    v->idata = p + FDB_PRIM_OFF_DATA;
    v->init = 0; v->slope = 0; v->nbits = 64; v->sign = 1;
  }
}

__device__ __forceinline__ int64_t d_inner_at(const DVec* v, int i) {
  switch (v->nbits) {
    case 32: return d_i32(v->idata + 4 * (size_t)i);
    case 16: { int32_t x = (int16_t)d_u16(v->idata + 2 * (size_t)i);
               return v->sign ? x : (x & 0xffff); }
    case 8:  { int32_t x = (int8_t)v->idata[i];
               return v->sign ? x : (x & 0xff); }
    case 4:  return (v->idata[i >> 1] >> ((i & 1) * 4)) & 0x0f;
    case 2:  return (v->idata[i >> 2] >> ((i & 3) * 2)) & 0x03;
  }
  return 0;
}

// wave-local LDS ordering only: s_waitcnt lgkmcnt(0) — unlike s_waitcnt(0)
// this does NOT drain vmcnt, so outstanding global loads/stores/atomics keep
// flying across the fence
__device__ __forceinline__ void d_wait_lds() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}

__device__ __forceinline__ int64_t d_lv_at(const DVec* v, int i) {
  if (v->wf == FDB_WF_DDV) return v->init + (int64_t)v->slope * i + d_inner_at(v, i);
  if (v->wf == FDB_WF_DDV_CONST) return v->init + (int64_t)v->slope * i;
  return d_i64(v->idata + 8 * (size_t)i);
}

__device__ __forceinline__ double d_dv_at(const DVec* v, int i) {
  if (v->wf == FDB_WF_PRIM64) return d_f64(v->idata + 8 * (size_t)i);
  return (double)d_lv_at(v, i);
}

// one-load header open: vectors are 64-B aligned in the device blob (padded
// tail), so a single 32-byte read covers every header field of all three
// scalar layouts — replaces d_vec_open's ~6 dependent global loads with one
// round trip. first_val (optional) returns element 0 (ts0 for a timestamp
// vector) parsed from the same 32 bytes.
__device__ __forceinline__ void d_vec_open_wide(const uint8_t* p, DVec* v,
                                                int64_t* first_val) {
  uint32_t h[8];
  memcpy(h, p, 32);
  const uint16_t wf = (uint16_t)(h[1] & 0xffff);
  v->wf = wf;
  v->dropped = ((h[1] >> 16) & FDB_DROP_MASK) != 0;
  if (wf == FDB_WF_DDV) {
    v->init = (int64_t)(((uint64_t)h[3] << 32) | h[2]);
    v->slope = (int32_t)h[4];
    const uint32_t w6 = h[6];                  // inner bytes 4..7
    v->nbits = (uint8_t)((w6 >> 16) & FDB_NBITS_MASK);
    v->sign = ((w6 >> 16) & FDB_SIGN_MASK) != 0;
    const int bitShift = (int)(w6 >> 24) & 0x3f;
    v->idata = p + FDB_DDV_OFF_INNER + FDB_PRIM_OFF_DATA;
    const int numBytes = (int)h[5];            // inner length word
    v->n = ((numBytes - 4) * 8 + (bitShift != 0 ? bitShift - 8 : 0)) / v->nbits;
    if (first_val) {
      int64_t in0;                             // inner element 0 is in h[7]
      switch (v->nbits) {
        case 32: in0 = (int32_t)h[7]; break;
        case 16: { int32_t x = (int16_t)(h[7] & 0xffff);
                   in0 = v->sign ? x : (x & 0xffff); } break;
        case 8:  { int32_t x = (int8_t)(h[7] & 0xff);
                   in0 = v->sign ? x : (x & 0xff); } break;
        case 4:  in0 = h[7] & 0x0f; break;
        default: in0 = h[7] & 0x03; break;
      }
      *first_val = v->init + in0;
    }
  } else if (wf == FDB_WF_DDV_CONST) {
    v->n = (int32_t)h[2];
    v->init = (int64_t)(((uint64_t)h[4] << 32) | h[3]);
    v->slope = (int32_t)h[5];
    v->idata = nullptr; v->nbits = 0; v->sign = 0;
    if (first_val) *first_val = v->init;
  } else {
    v->n = ((int)h[0] - 4) / 8;
    v->idata = p + FDB_PRIM_OFF_DATA;
    v->init = 0; v->slope = 0; v->nbits = 64; v->sign = 1;
    if (first_val) *first_val = (int64_t)(((uint64_t)h[3] << 32) | h[2]);
  }
}

// wave-wide inclusive prefix sums (64 lanes)
__device__ __forceinline__ double wave_incl_scan(double x, int lane) {
  for (int off = 1; off < 64; off <<= 1) {
    double t = __shfl_up(x, off);
    if (lane >= off) x += t;
  }
  return x;
}
__device__ __forceinline__ int wave_incl_scan_i(int x, int lane) {
  for (int off = 1; off < 64; off <<= 1) {
    int t = __shfl_up(x, off);
    if (lane >= off) x += t;
  }
  return x;
}

// vectorized decode: whole chunk into LDS, wide loads, high ILP.
// (element semantics identical to d_lv_at/d_dv_at, which remain the
//  reference implementations / fallback)
template <bool AS_DOUBLE>
__device__ inline void d_decode_chunk(const DVec& v, int n, int64_t* tout,
                                      double* dout, int lane) {
  if (v.wf == FDB_WF_DDV_CONST) {
    for (int i = lane; i < n; i += 64) {
      int64_t x = v.init + (int64_t)v.slope * i;
      if (AS_DOUBLE) dout[i] = (double)x; else tout[i] = x;
    }
    return;
  }
  if (v.wf == FDB_WF_PRIM64) {
    // raw 64-bit payload, 8-byte aligned (+8 from a 64B-aligned base)
    for (int i = lane; i < n; i += 64) {
      if (AS_DOUBLE) dout[i] = d_f64(v.idata + 8 * (size_t)i);
      else           tout[i] = d_i64(v.idata + 8 * (size_t)i);
    }
    return;
  }
  // packed DDV inner data starts at +28 (4-byte aligned only)
  if (v.nbits == 16) {
    for (int i0 = 4 * lane; i0 < n; i0 += 256) {
      uint32_t lo = d_u32(v.idata + 2 * (size_t)i0);
      uint32_t hi = d_u32(v.idata + 2 * (size_t)i0 + 4);
      uint64_t w = ((uint64_t)hi << 32) | lo;
      #pragma unroll
      for (int k = 0; k < 4; k++) {
        if (i0 + k < n) {
          int32_t d = (int32_t)(int16_t)(uint16_t)(w >> (16 * k));
          if (!v.sign) d &= 0xffff;
          int64_t x = v.init + (int64_t)v.slope * (i0 + k) + d;
          if (AS_DOUBLE) dout[i0 + k] = (double)x; else tout[i0 + k] = x;
        }
      }
    }
    return;
  }
  if (v.nbits == 8) {
    for (int i0 = 8 * lane; i0 < n; i0 += 512) {
      uint32_t lo = d_u32(v.idata + (size_t)i0);
      uint32_t hi = d_u32(v.idata + (size_t)i0 + 4);
      uint64_t w = ((uint64_t)hi << 32) | lo;
      #pragma unroll
      for (int k = 0; k < 8; k++) {
        if (i0 + k < n) {
          int32_t d = (int32_t)(int8_t)(uint8_t)(w >> (8 * k));
          if (!v.sign) d &= 0xff;
          int64_t x = v.init + (int64_t)v.slope * (i0 + k) + d;
          if (AS_DOUBLE) dout[i0 + k] = (double)x; else tout[i0 + k] = x;
        }
      }
    }
    return;
  }
  if (v.nbits == 32) {
    for (int i0 = 2 * lane; i0 < n; i0 += 128) {
      #pragma unroll
      for (int k = 0; k < 2; k++) {
        if (i0 + k < n) {
          int64_t x = v.init + (int64_t)v.slope * (i0 + k)
                    + d_i32(v.idata + 4 * (size_t)(i0 + k));
          if (AS_DOUBLE) dout[i0 + k] = (double)x; else tout[i0 + k] = x;
        }
      }
    }
    return;
  }
  // nbits 2/4 fallback (rare)
  for (int i = lane; i < n; i += 64) {
    int64_t x = v.init + (int64_t)v.slope * i + d_inner_at(&v, i);
    if (AS_DOUBLE) dout[i] = (double)x; else tout[i] = x;
  }
}

// i32-offset decode variant: timestamps relative to the chunk's first ts.
// Caller guarantees the chunk's time span fits i32 (upload-time check).
__device__ inline void d_decode_ts_offsets(const DVec& v, int n, int64_t ts0,
                                           int32_t* tout, int lane) {
  // all arithmetic in i32: the upload guard pins the chunk's span (and so
  // base + slope*i + inner, = ts_i - ts0) inside i32, and i32 wrap-around
  // equals the i64-then-truncate result whenever the true value fits
  if (v.wf == FDB_WF_DDV_CONST) {
    const int32_t cbase = (int32_t)(v.init - ts0);
    for (int i = lane; i < n; i += 64)
      tout[i] = (int32_t)((uint32_t)v.slope * (uint32_t)i + (uint32_t)cbase);
    return;
  }
  if (v.wf == FDB_WF_PRIM64) {
    for (int i = lane; i < n; i += 64)
      tout[i] = (int32_t)(d_i64(v.idata + 8 * (size_t)i) - ts0);
    return;
  }
  const int32_t base = (int32_t)(v.init - ts0);
  if (v.nbits == 16) {
    for (int i0 = 4 * lane; i0 < n; i0 += 256) {
      uint32_t lo = d_u32(v.idata + 2 * (size_t)i0);
      uint32_t hi = d_u32(v.idata + 2 * (size_t)i0 + 4);
      uint64_t w = ((uint64_t)hi << 32) | lo;
      #pragma unroll
      for (int k = 0; k < 4; k++) {
        if (i0 + k < n) {
          int32_t d = (int32_t)(int16_t)(uint16_t)(w >> (16 * k));
          if (!v.sign) d &= 0xffff;
          tout[i0 + k] = (int32_t)((uint32_t)base
              + (uint32_t)v.slope * (uint32_t)(i0 + k) + (uint32_t)d);
        }
      }
    }
    return;
  }
  if (v.nbits == 32) {
    for (int i0 = 2 * lane; i0 < n; i0 += 128) {
      #pragma unroll
      for (int k = 0; k < 2; k++)
        if (i0 + k < n)
          tout[i0 + k] = (int32_t)((uint32_t)base
              + (uint32_t)v.slope * (uint32_t)(i0 + k)
              + (uint32_t)d_i32(v.idata + 4 * (size_t)(i0 + k)));
    }
    return;
  }
  for (int i = lane; i < n; i += 64)
    tout[i] = (int32_t)((uint32_t)base + (uint32_t)v.slope * (uint32_t)i
                        + (uint32_t)d_inner_at(&v, i));
}

// Correctly-rounded x/1000 in 3 ops (mul + 2 fma, Markstein fixup) instead of
// the ~10-instruction v_div_scale/v_div_fmas/v_div_fixup sequence. EXACTLY
// equal to RN(x/1000) for every integer |x| <= 2^33 — verified exhaustively
// (8.59e9 cases, 0 mismatches; the naive q0 = x*(1/1000) alone differs on 13%
// of them). Callers pass integer millisecond DIFFERENCES bounded by the
// window length; the engine rejects windows > 2^33 ms (~99 days) up front.
__device__ __forceinline__ double d_div1000(double x) {
  constexpr double r = 1.0 / 1000.0;                // RN reciprocal
  double q0 = x * r;
  double e = __builtin_fma(-1000.0, q0, x);         // exact residual
  return __builtin_fma(e, r, q0);
}

// extrapolatedRate (RateFunctions.scala:72-111) — the oracle's EXACT
// operation sequence. Do not reassociate or replace the divisions: the
// durationToZero/threshold comparisons are discontinuous and integer counter
// data makes exact rational ties (v1/delta == 1.1/(numSamples-1)) common, so
// the branch taken must follow the reference's own FP rounding. (The /1000
// sites use d_div1000 — bit-identical on the engine's ms-difference domain.)
__device__ inline double d_extrapolated_rate(int64_t windowStart, int64_t windowEnd,
                                             int numSamples,
                                             int64_t t1, double v1, int64_t t2, double v2,
                                             bool isCounter, bool isRate) {
  double durationToStart = d_div1000((double)(t1 - windowStart));
  double durationToEnd = d_div1000((double)(windowEnd - t2));
  double sampledInterval = d_div1000((double)(t2 - t1));
  double avgDur = sampledInterval / ((double)numSamples - 1);
  double delta = v2 - v1;
  if (isCounter && delta > 0 && v1 >= 0) {
    double durationToZero = sampledInterval * (v1 / delta);
    if (durationToZero < durationToStart) durationToStart = durationToZero;
  }
  double thresh = avgDur * 1.1;
  double ext = sampledInterval;
  ext += (durationToStart < thresh) ? durationToStart : avgDur / 2;
  ext += (durationToEnd < thresh) ? durationToEnd : avgDur / 2;
  double scaledDelta = delta * (ext / sampledInterval);
  return isRate ? (scaledDelta / (double)(windowEnd - windowStart) * 1000.0) : scaledDelta;
}

// i32-difference variant of d_extrapolated_rate for the fast path: all three
// millisecond gaps fit i32 when the window does (caller checks), so each
// i64 subtract + 4-instruction i64→f64 convert becomes a truncate + single
// v_cvt_f64_i32; the window duration is wave-uniform and passed pre-converted.
// Every operation after the converts matches d_extrapolated_rate exactly
// (same values, same rounding) — results are bit-identical.
__device__ inline double d_extrapolated_rate_i32(int32_t toStart_ms, int32_t toEnd_ms,
                                                 int numSamples, int32_t sampled_ms,
                                                 double dur_ms, double v1, double v2,
                                                 bool isCounter, bool isRate) {
  double durationToStart = d_div1000((double)toStart_ms);
  double durationToEnd = d_div1000((double)toEnd_ms);
  double sampledInterval = d_div1000((double)sampled_ms);
  double avgDur = sampledInterval / ((double)numSamples - 1);
  double delta = v2 - v1;
  if (isCounter && delta > 0 && v1 >= 0) {
    double durationToZero = sampledInterval * (v1 / delta);
    if (durationToZero < durationToStart) durationToStart = durationToZero;
  }
  double thresh = avgDur * 1.1;
  double ext = sampledInterval;
  ext += (durationToStart < thresh) ? durationToStart : avgDur / 2;
  ext += (durationToEnd < thresh) ? durationToEnd : avgDur / 2;
  double scaledDelta = delta * (ext / sampledInterval);
  return isRate ? (scaledDelta / dur_ms * 1000.0) : scaledDelta;
}

// NaN-aware f64 atomic min/max via CAS (group aggregation; RowAggregator semantics)
__device__ inline void atomic_min_max_f64(double* addr, double val, bool is_min) {
  unsigned long long* a = (unsigned long long*)addr;
  unsigned long long old = *a, assumed;
  do {
    assumed = old;
    double cur = __longlong_as_double((long long)assumed);
    double nw = isnan(cur) ? val : (is_min ? fmin(cur, val) : fmax(cur, val));
    if (!isnan(cur) && nw == cur) return;
    old = atomicCAS(a, assumed, (unsigned long long)__double_as_longlong(nw));
  } while (old != assumed);
}

// floor(a/b) clamped into the window index range [-1, hi+1], for b>0:
// double reciprocal + one-step integer fixup (i64 division is software-
// emulated and costs hundreds of cycles). Outside [-1, hi+1] the exact value
// is irrelevant — callers clamp ranges to [0, hi) — so the guards also keep
// the fixup products within i64 (callers ensure b <= 2^31).
__device__ __forceinline__ int fdiv_floor_win(int64_t a, int64_t b, double inv_b,
                                              int hi) {
  double g = (double)a * inv_b;
  if (g < -1.0) return -1;
  if (g > (double)hi + 1.0) return hi + 1;
  int64_t w = (int64_t)__builtin_floor(g);
  // residual a - w*b via fma: |residual| < 2b << 2^53, so the single-rounded
  // fma result is exact — replaces two software i64 multiplies
  double r = __builtin_fma(-(double)w, (double)b, (double)a);
  w += (r >= (double)b);
  w -= (r < 0.0);
  return (int)w;
}

// fdiv_floor_win widened: clamp low bound is a parameter and the exact
// "residual nonzero" bit comes back, so ONE division yields both inversion
// boundaries when b | window: f = floor(a/b), ci = f + (a mod b != 0) [ceil],
// di = f + window/b. Outside [lo_clamp, hi+1] the value is only ever compared
// against in-range window indexes (callers take min/max), so the clamped
// return and a pessimistic nz are safe; the 1.0 gap between adjacent
// quotients dwarfs the ~1e-7 absolute error of g, so the clamp tests cannot
// fire for a true quotient inside the range.
__device__ __forceinline__ int fdiv_floor_rem(int64_t a, int64_t b, double inv_b,
                                              int hi, int lo_clamp, bool* nz) {
  double g = (double)a * inv_b;
  if (g < (double)lo_clamp) { *nz = true; return lo_clamp; }
  if (g > (double)hi + 1.0) { *nz = true; return hi + 1; }
  int64_t w = (int64_t)__builtin_floor(g);
  double r = __builtin_fma(-(double)w, (double)b, (double)a);
  const bool up = (r >= (double)b), dn = (r < 0.0);
  w += up;
  w -= dn;
  double rr = up ? r - (double)b : (dn ? r + (double)b : r);
  *nz = (rr != 0.0);
  return (int)w;
}

// wave-wide inclusive prefix sum over i64 (bucket-cumulative reconstruction)
__device__ __forceinline__ int64_t wave_incl_scan_i64(int64_t x, int lane) {
  for (int off = 1; off < 64; off <<= 1) {
    int64_t t = __shfl_up(x, off);
    if (lane >= off) x += t;
  }
  return x;
}

// --- wave-staged element stream -------------------------------------------
// One coalesced dword-per-lane load stages 256 B of an element's NibblePack
// stream into the wave's registers; all subsequent byte/u16/u64 reads are
// register shuffles (ds_bpermute) instead of dependent global loads. The
// stage for element e+1 is issued before element e is parsed, so the load
// latency overlaps a full element's parse work. Elements longer than the
// staged range (elen+shift > 252, i.e. huge sections) fall back to direct
// global reads; the blob is tail-padded 256 B at upload so staging never
// faults.
__device__ __forceinline__ uint32_t estream_stage(const uint8_t* p, int lane) {
  const uint8_t* base = (const uint8_t*)((uintptr_t)p & ~(uintptr_t)3);
  uint32_t d;
  memcpy(&d, base + lane * 4, 4);
  return d;
}
// `staged` must be WAVE-UNIFORM and every call site must have the full wave
// active: __shfl is ds_bpermute, and a source lane that is inactive (or a
// divergent caller) yields undefined data.
__device__ __forceinline__ uint32_t estream_byte(bool staged, uint32_t buf,
                                                 int shift, const uint8_t* g,
                                                 int k) {
  if (staged) {
    int kk = k + shift;
    uint32_t d = __shfl(buf, kk >> 2);
    return (d >> ((kk & 3) * 8)) & 0xff;
  }
  return g[k];
}
__device__ __forceinline__ uint32_t estream_u16(bool staged, uint32_t buf,
                                                int shift, const uint8_t* g,
                                                int k) {
  return estream_byte(staged, buf, shift, g, k) |
         (estream_byte(staged, buf, shift, g, k + 1) << 8);
}
// wave-UNIFORM byte read from the staged window: kk>>2 is the same for every
// lane, so v_readlane (scalar result, no LDS pipe) replaces ds_bpermute —
// the header walks of the NibblePack parsers are built from these
__device__ __forceinline__ uint32_t estream_byte_uni(bool staged, uint32_t buf,
                                                     int shift, const uint8_t* g,
                                                     int k) {
  if (staged) {
    int kk = k + shift;
    uint32_t d = __builtin_amdgcn_readlane(buf, kk >> 2);
    return (d >> ((kk & 3) * 8)) & 0xff;
  }
  return g[k];
}

// little-endian 8-byte window at offset k (per-lane k; bpermute shuffles)
__device__ __forceinline__ uint64_t estream_w64(bool staged, uint32_t buf,
                                                int shift, const uint8_t* g,
                                                int k) {
  if (staged) {
    int kk = k + shift;
    int dw = kk >> 2;
    uint64_t a = __shfl(buf, dw);
    uint64_t b = __shfl(buf, dw + 1);
    uint64_t c = __shfl(buf, dw + 2);
    int sh = (kk & 3) * 8;
    uint64_t w = (a | (b << 32)) >> sh;
    if (sh) w |= c << (64 - sh);
    return w;
  }
  uint64_t w;
  memcpy(&w, g + k, 8);
  return w;
}


#endif  // FDB_SCAN_COMMON_H
