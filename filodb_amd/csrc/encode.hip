// GPU ingest-side chunk encoder (SURVEY §8f "ingest-side GPU encode").
//
// One WAVE per chunk: reproduces the host encoder's frozen scalar vectors
// BYTE-EXACTLY (chunk_builder.cpp encode_timestamps/encode_doubles, which
// restate DeltaDeltaVector.scala:63-135, IntBinaryVector.scala:52-177,
// DoubleVector.scala:86-96,457-476, LongBinaryVector.scala:333-341):
//
//   timestamps — DDV(approxConst=true): slope = (last-first)/(n-1) truncating
//     i64 division; all line deltas within i32 required; ±250 band (or all
//     zero) → const vector; else packed inner int vector with
//     minMaxToNbitsSigned nbits∈{2,4,8,16,32}; fallback raw i64.
//   doubles — DDV when every value is an integral double inside the Long
//     round-trip bounds (NaN is non-integral) and n > 2, else raw f64; the
//     counter drop bit is recomputed from the rows exactly like
//     DoubleCounterAppender.addData (NaN or a decrease vs the latest
//     preceding non-NaN value).
//
// Parallel shape: lanes stride the rows for the delta min/max and integral
// reductions; packing writes disjoint bytes (nbits 2/4 pack via shfl from
// the byte's 2-4 source lanes). Parity = byte equality with the host store
// (tests/test_gpu_encode.py) — the host encoder is itself pinned to the
// reference by the codec byte-golden tests.

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstring>
#include <cmath>

#include "chunk_format.h"
#include "scan_common.h"
#include "../../include/filodb_amd.h"

void fdb_set_error(const char* fmt, ...);         // chunk_builder.cpp
hipStream_t fdb_engine_stream(fdb_engine_t* e);   // engine.hip


__device__ __forceinline__ int32_t wave_min_i32(int32_t x) {
  for (int off = 32; off > 0; off >>= 1) x = min(x, __shfl_xor(x, off));
  return __builtin_amdgcn_readfirstlane(x);
}
__device__ __forceinline__ int32_t wave_max_i32(int32_t x) {
  for (int off = 32; off > 0; off >>= 1) x = max(x, __shfl_xor(x, off));
  return __builtin_amdgcn_readfirstlane(x);
}
__device__ __forceinline__ bool wave_all(bool p) {
  return __all(p ? 1 : 0);
}
__device__ __forceinline__ bool wave_any(bool p) {
  return __any(p ? 1 : 0);
}

// minMaxToNbitsSigned (IntBinaryVector.scala:161-177)
__device__ void d_nbits_signed(int32_t mn, int32_t mx, int* nbits, bool* sign) {
  if (mn >= 0 && mx < 4)            { *nbits = 2;  *sign = false; }
  else if (mn >= 0 && mx < 16)      { *nbits = 4;  *sign = false; }
  else if (mn >= -128 && mx <= 127) { *nbits = 8;  *sign = true;  }
  else if (mn >= 0 && mx < 256)     { *nbits = 8;  *sign = false; }
  else if (mn >= -32768 && mx <= 32767) { *nbits = 16; *sign = true; }
  else if (mn >= 0 && mx < 65536)   { *nbits = 16; *sign = false; }
  else                              { *nbits = 32; *sign = true;  }
}

struct LongSrc {          // rows as i64: ts directly, or integral doubles cast
  const int64_t* ts;
  const double* vals;
  bool from_doubles;
  __device__ int64_t at(int i) const {
    return from_doubles ? (int64_t)vals[i] : ts[i];
  }
};

// Writes one DDV (const or packed) vector at out; returns its total byte
// length, or -1 if the longs are not DDV-eligible. Wave-cooperative;
// all lanes return the same value. (__noinline__ keeps the kernel body
// small; the optimizer in this ROCm's clang-22 is fragile around this TU —
// see the enc_raw64 note)
__device__ __noinline__ int enc_ddv_longs(const LongSrc& src, int n, bool approx_const,
                             uint8_t* out, int lane) {
  if (n <= 2) return -1;                         // fromLongVector :67
  const int64_t first = src.at(0);
  const int64_t slope64 = (src.at(n - 1) - first) / (int64_t)(n - 1);
  if (slope64 >= INT32_MAX || slope64 <= INT32_MIN) return -1;
  const int32_t slope = (int32_t)slope64;
  int32_t mn = INT32_MAX, mx = INT32_MIN;
  bool fits = true;
  for (int i = 1 + lane; i < n; i += 64) {
    int64_t d = src.at(i) - (first + slope64 * i);
    if (d > INT32_MAX || d < INT32_MIN) fits = false;
    else {
      if ((int32_t)d > mx) mx = (int32_t)d;
      if ((int32_t)d < mn) mn = (int32_t)d;
    }
  }
  if (!wave_all(fits)) return -1;
  mn = wave_min_i32(mn);
  mx = wave_max_i32(mx);
  int nbits; bool sign;
  d_nbits_signed(mn, mx, &nbits, &sign);

  if ((mn == 0 && mx == 0) ||
      (approx_const && mn >= -FDB_DDV_MAX_APPROX_DELTA &&
       mx <= FDB_DDV_MAX_APPROX_DELTA)) {
    if (lane == 0) {                             // const DDV (:89-106)
      uint32_t h0 = 20, h1 = FDB_WF_DDV_CONST, nn = (uint32_t)n;
      memcpy(out + 0, &h0, 4);
      memcpy(out + 4, &h1, 4);
      memcpy(out + FDB_DDVC_OFF_NELEM, &nn, 4);
      memcpy(out + FDB_DDVC_OFF_INIT, &first, 8);
      memcpy(out + FDB_DDVC_OFF_SLOPE, &slope, 4);
    }
    return FDB_DDVC_BYTES;
  }

  // packed DDV: element i's delta vs the line (element 0's is 0)
  const int data = FDB_DDV_OFF_INNER + FDB_PRIM_OFF_DATA;
  const int dataBytes = (n * nbits + 7) / 8;
  const int bitShift = (n * nbits) % 8;
  if (lane == 0) {
    uint32_t wf = FDB_WF_DDV;
    memcpy(out + 4, &wf, 4);
    memcpy(out + FDB_DDV_OFF_INIT, &first, 8);
    memcpy(out + FDB_DDV_OFF_SLOPE, &slope, 4);
    uint16_t iwf = FDB_WF_INT_NOMASK;
    memcpy(out + FDB_DDV_OFF_INNER + 4, &iwf, 2);
    out[FDB_DDV_OFF_INNER + 6] =
        (uint8_t)((nbits & FDB_NBITS_MASK) | (sign ? FDB_SIGN_MASK : 0));
    out[FDB_DDV_OFF_INNER + 7] = (uint8_t)bitShift;
    uint32_t ilen = (uint32_t)(4 + dataBytes);
    memcpy(out + FDB_DDV_OFF_INNER, &ilen, 4);
    uint32_t olen = (uint32_t)(data + dataBytes - 4);
    memcpy(out, &olen, 4);
  }
  for (int base = 0; base < n; base += 64) {
    const int i = base + lane;
    const int32_t v = (i < n) ? (int32_t)(src.at(i) - (first + slope64 * i))
                              : 0;
    switch (nbits) {
      case 32: if (i < n) memcpy(out + data + 4 * (size_t)i, &v, 4); break;
      case 16: { uint16_t h = (uint16_t)(int16_t)v;
                 if (i < n) memcpy(out + data + 2 * (size_t)i, &h, 2); } break;
      case 8:  if (i < n) out[data + i] = (uint8_t)(int8_t)v; break;
      case 4: {                        // low nibble first (IntBinaryVector :84-105)
        uint32_t nx = __shfl((uint32_t)v, lane + 1);
        if (i < n && (lane & 1) == 0) {
          uint8_t byte = (uint8_t)((uint32_t)v & 0x0f);
          if (i + 1 < n) byte |= (uint8_t)((nx & 0x0f) << 4);
          out[data + i / 2] = byte;
        }
      } break;
      default: {                       // nbits == 2
        uint32_t n1 = __shfl((uint32_t)v, lane + 1);
        uint32_t n2 = __shfl((uint32_t)v, lane + 2);
        uint32_t n3 = __shfl((uint32_t)v, lane + 3);
        if (i < n && (lane & 3) == 0) {
          uint8_t byte = (uint8_t)((uint32_t)v & 0x03);
          if (i + 1 < n) byte |= (uint8_t)((n1 & 3) << 2);
          if (i + 2 < n) byte |= (uint8_t)((n2 & 3) << 4);
          if (i + 3 < n) byte |= (uint8_t)((n3 & 3) << 6);
          out[data + i / 4] = byte;
        }
      } break;
    }
  }
  return data + dataBytes;
}

// raw 64-bit primitive vector with optional drop bit
__device__ int enc_raw64(const int64_t* ts, const double* vals, int n,
                         bool drop, uint8_t* out, int lane) {
  if (lane == 0) {
    uint32_t len = (uint32_t)(4 + 8 * n);
    memcpy(out, &len, 4);
    uint16_t wf = FDB_WF_PRIM64;
    memcpy(out + 4, &wf, 2);
    uint16_t w6 = (uint16_t)((64 | FDB_SIGN_MASK) | (drop ? FDB_DROP_MASK : 0));
    memcpy(out + 6, &w6, 2);
  }
  // single hoisted source pointer: the per-iteration ts/vals branch form
  // reliably segfaults clang-22's -O3 optimization pipeline on gfx950
  const uint8_t* src = ts ? (const uint8_t*)ts : (const uint8_t*)vals;
  for (int i = lane; i < n; i += 64) {
    uint64_t w;
    memcpy(&w, src + 8 * (size_t)i, 8);
    memcpy(out + FDB_PRIM_OFF_DATA + 8 * (size_t)i, &w, 8);
  }
  return FDB_PRIM_OFF_DATA + 8 * n;
}


// one wave per chunk; out offsets precomputed host-side (padded upper bounds)
__global__ __launch_bounds__(256)
void encode_kernel(const int64_t* __restrict__ ts,
                   const double* __restrict__ vals,
                   const int64_t* __restrict__ row_offs,   // [nchunks+1]
                   int nchunks, int col_kind,
                   uint8_t* __restrict__ out,
                   const int64_t* __restrict__ ts_off,
                   const int64_t* __restrict__ val_off,
                   int32_t* __restrict__ ts_len,
                   int32_t* __restrict__ val_len) {
  const int wave = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  for (int c = blockIdx.x * 4 + wave; c < nchunks; c += gridDim.x * 4) {
    const int64_t s0 = row_offs[c];
    const int n = (int)(row_offs[c + 1] - s0);
    if (n < 1) { if (lane == 0) { ts_len[c] = 0; val_len[c] = 0; } continue; }
    const int64_t* cts = ts + s0;
    const double* cvals = vals + s0;

    // ---- timestamps: approx-const DDV else raw i64 -------------------------
    LongSrc tsrc{cts, nullptr, false};
    int tl = enc_ddv_longs(tsrc, n, /*approx_const=*/true, out + ts_off[c], lane);
    if (tl < 0) tl = enc_raw64(cts, nullptr, n, false, out + ts_off[c], lane);
    if (lane == 0) ts_len[c] = tl;

    // ---- values ------------------------------------------------------------
    // counter drop flag: NaN anywhere, or a non-NaN value below the latest
    // preceding non-NaN value (DoubleCounterAppender.addData) — computed with
    // a "latest valid" wave scan carried across 64-row windows
    // "latest preceding non-NaN" is just the previous VALID element in
    // sequence order, so per 64-row window: ballot the valid lanes, each
    // valid lane compares against the nearest valid lane below it (one
    // shfl), the lowest valid lane against the carried last-valid value.
    bool drop = false;
    if (col_kind == FDB_COL_COUNTER) {
      double carry = -1.7976931348623157e308;    // Double.MinValue start
      bool any_bad = false;
      for (int base = 0; base < n; base += 64) {
        const int i = base + lane;
        const double v = (i < n) ? cvals[i] : NAN;
        const bool valid = (i < n) && !isnan(v);
        if (i < n && !valid) any_bad = true;     // NaN ⇒ drop
        const unsigned long long mval = __ballot(valid ? 1 : 0);
        const unsigned long long below =
            mval & ((lane ? ~0ULL >> (64 - lane) : 0ULL));
        const int prev_lane = below ? 63 - __clzll(below) : 0;
        const double prev = __shfl(v, prev_lane);
        if (valid && (below ? v < prev : v < carry)) any_bad = true;
        if (mval) carry = __shfl(v, 63 - __clzll(mval));
      }
      drop = wave_any(any_bad);
    }

    // integral check (Long round-trip bounds; NaN non-integral)
    bool integral = true;
    for (int i = lane; i < n; i += 64) {
      const double d = cvals[i];
      if (d > 9.2233720368547758e18 || d < -9.2233720368547758e18 ||
          rint(d) != d)
        integral = false;
    }
    integral = wave_all(integral);

    int vl = -1;
    if (integral && n > 2) {
      LongSrc vsrc{nullptr, cvals, true};
      vl = enc_ddv_longs(vsrc, n, /*approx_const=*/false, out + val_off[c], lane);
      if (vl > 0 && drop && lane == 0) {         // DoubleCounterAppender.optimize
        uint16_t w;
        memcpy(&w, out + val_off[c] + 6, 2);
        w |= FDB_DROP_MASK;
        memcpy(out + val_off[c] + 6, &w, 2);
      }
    }
    if (vl < 0) vl = enc_raw64(nullptr, cvals, n, drop, out + val_off[c], lane);
    if (lane == 0) val_len[c] = vl;
  }
}

// Host entry: encodes num_chunks scalar chunks on the GPU. ts/vals are
// concatenated rows (host); row_offs[c]..row_offs[c+1] delimit chunk c.
// Outputs land in out (host, caller-sized): chunk c's frozen timestamp vector
// at out_ts_off[c] (length out_ts_len[c]) and value vector at out_val_off[c].
// Offsets are assigned inside (64-B aligned, padded upper bounds).
extern "C" int32_t fdb_gpu_encode_chunks(fdb_engine_t* e,
                                         const int64_t* ts, const double* vals,
                                         const int64_t* row_offs,
                                         int32_t num_chunks, int32_t col_kind,
                                         uint8_t* out, int64_t out_cap,
                                         int64_t* out_ts_off, int64_t* out_val_off,
                                         int32_t* out_ts_len, int32_t* out_val_len) {
  if (num_chunks < 1 || !ts || !vals || !row_offs) {
    fdb_set_error("gpu_encode: bad args");
    return FDB_ERR_BADARG;
  }
  if (col_kind == FDB_COL_HIST) {
    fdb_set_error("gpu_encode: histogram columns encode host-side");
    return FDB_ERR_BADARG;
  }
  for (int c = 0; c < num_chunks; c++) {
    if (row_offs[c + 1] < row_offs[c] || row_offs[c] < 0) {
      fdb_set_error("gpu_encode: row_offs must be nonnegative and nondecreasing");
      return FDB_ERR_BADARG;
    }
  }
  const int64_t nrows = row_offs[num_chunks];
  // padded upper bounds: ts ≤ max(DDV 28+4n, raw 8+8n)+align; val ≤ 12+8n
  int64_t need = 0;
  for (int c = 0; c < num_chunks; c++) {
    const int64_t n = row_offs[c + 1] - row_offs[c];
    out_ts_off[c] = need;
    need += ((n * 8 + 64 + 63) & ~63LL);
    out_val_off[c] = need;
    need += ((n * 8 + 64 + 63) & ~63LL);
  }
  if (need > out_cap) {
    fdb_set_error("gpu_encode: need %lld bytes, cap %lld",
                  (long long)need, (long long)out_cap);
    return FDB_ERR_BADARG;
  }
  hipStream_t stream = fdb_engine_stream(e);
  int64_t *dts = nullptr, *droffs = nullptr, *dtoff = nullptr, *dvoff = nullptr;
  double* dvals = nullptr;
  uint8_t* dout = nullptr;
  int32_t *dtlen = nullptr, *dvlen = nullptr;
  int32_t rc = FDB_ERR;
  if (hipMalloc(&dts, (size_t)nrows * 8) != hipSuccess) goto done;
  if (hipMalloc(&dvals, (size_t)nrows * 8) != hipSuccess) goto done;
  if (hipMalloc(&droffs, ((size_t)num_chunks + 1) * 8) != hipSuccess) goto done;
  if (hipMalloc(&dtoff, (size_t)num_chunks * 8) != hipSuccess) goto done;
  if (hipMalloc(&dvoff, (size_t)num_chunks * 8) != hipSuccess) goto done;
  if (hipMalloc(&dtlen, (size_t)num_chunks * 4) != hipSuccess) goto done;
  if (hipMalloc(&dvlen, (size_t)num_chunks * 4) != hipSuccess) goto done;
  if (hipMalloc(&dout, (size_t)need) != hipSuccess) goto done;
  if (hipMemset(dout, 0, (size_t)need) != hipSuccess) goto done;
  if (hipMemcpy(dts, ts, (size_t)nrows * 8, hipMemcpyHostToDevice) != hipSuccess) goto done;
  if (hipMemcpy(dvals, vals, (size_t)nrows * 8, hipMemcpyHostToDevice) != hipSuccess) goto done;
  if (hipMemcpy(droffs, row_offs, ((size_t)num_chunks + 1) * 8, hipMemcpyHostToDevice) != hipSuccess) goto done;
  if (hipMemcpy(dtoff, out_ts_off, (size_t)num_chunks * 8, hipMemcpyHostToDevice) != hipSuccess) goto done;
  if (hipMemcpy(dvoff, out_val_off, (size_t)num_chunks * 8, hipMemcpyHostToDevice) != hipSuccess) goto done;
  {
    int grid = (num_chunks + 3) / 4;
    if (grid > 16384) grid = 16384;
    hipLaunchKernelGGL(encode_kernel, dim3(grid), dim3(256), 0, stream,
                       dts, dvals, droffs, num_chunks, col_kind, dout,
                       dtoff, dvoff, dtlen, dvlen);
    if (hipGetLastError() != hipSuccess) {
      fdb_set_error("encode_kernel launch failed");
      goto done;
    }
  }
  if (hipStreamSynchronize(stream) != hipSuccess) goto done;
  if (hipMemcpy(out, dout, (size_t)need, hipMemcpyDeviceToHost) != hipSuccess) goto done;
  if (hipMemcpy(out_ts_len, dtlen, (size_t)num_chunks * 4, hipMemcpyDeviceToHost) != hipSuccess) goto done;
  if (hipMemcpy(out_val_len, dvlen, (size_t)num_chunks * 4, hipMemcpyDeviceToHost) != hipSuccess) goto done;
  rc = FDB_OK;
done:
  if (rc != FDB_OK) fdb_set_error("gpu_encode: device error");
  (void)hipFree(dts); (void)hipFree(dvals); (void)hipFree(droffs);
  (void)hipFree(dtoff); (void)hipFree(dvoff); (void)hipFree(dtlen);
  (void)hipFree(dvlen); (void)hipFree(dout);
  return rc;
}
