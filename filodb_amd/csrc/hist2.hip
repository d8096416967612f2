// Histogram scan v2: two-cursor monotonic walk over the sect-delta streams.
//
// Round 1 streamed every element once, holding per-bucket window-start values
// in a ring of active windows — which capped window/step at 22 and chunks at
// 4, and serialized on the per-element header parse (43.7 ms for the
// BASELINE config #4 workload; profiles/README.md). v2 removes the ring:
//
//   rate(hist[w]) per window needs only the FIRST element >= wStart and the
//   LAST element <= wEnd (HistogramRateFunctionBase, RateFunctions.scala:
//   330-400). Windows are processed in order by ONE wave per series with two
//   monotone cursors — S tracks the window-start element, E the window-end
//   element. A cursor hops elements by their u16 length prefix and fully
//   decodes an element only when it stops on it (or at section bases, since
//   sect-delta diffs are against the section's first element —
//   HistogramVector.scala:491-545) — so the dependent header-parse chain runs
//   ~twice per element worst case, with the two cursors' chains independent
//   and overlapping, at far higher occupancy (no 12 KB ring in LDS).
//
// Unbounded: chunks per series, window/step ratio, num_windows. Per-chunk
// rows <= 400 (the per-cursor LDS timestamp buffer; the reference's own
// chunk cap). Corrections: TypeDrop sections add apply(e-1) per bucket
// (Section.scala:17-24 type byte); chunk-boundary drops use
// Histogram.compare's top-bucket-down order (Histogram.scala:204-214) —
// same rules and per-window correction base as v1 (parity-green).

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstring>
#include <cmath>

#include "chunk_format.h"
#include "scan_common.h"
#include "../../include/filodb_amd.h"

void fdb_set_error(const char* fmt, ...);   // chunk_builder.cpp

#define H2_ROWS 400
#define H2_WAVES 4

// parse one element's per-lane bucket delta from its NibblePack stream.
// ep = element start (at the u16 length); wave-uniform. All lanes must be
// active (estream shuffles). Returns the lane's (<<trailing-zeroes) delta.
// The _buf form takes a pre-staged 256-B register window so two elements'
// stage loads can be issued together and their parse chains interleaved.
template <bool EST>
__device__ __forceinline__ int64_t h2_parse_est(const uint8_t* ep, int elen,
                                                int nb, int b, bool live,
                                                int lane, uint32_t ebuf,
                                                int eshift) {
  // 8-value group headers, walked serially (wave-uniform). EST is a
  // compile-time constant: with it true the walk is pure readlane + scalar
  // arithmetic (SGPR chain, no exec-mask churn); the runtime est ? : form
  // made every access an if-converted dual path whose global-load arm
  // poisoned the uniformity analysis and kept the whole walk on the VALU.
  const int my_group = b >> 3;
  int off = 0, gOff = 0, gBits = 0, gTrail = 0;
  uint32_t gMask = 0;
  for (int g = 0; g * 8 < nb; g++) {
    uint32_t mask = estream_byte_uni(EST, ebuf, eshift, ep, 2 + off);
    int numBits = 0, trail = 0, glen;
    if (mask == 0) {
      glen = 1;
    } else {
      int widths = (int)estream_byte_uni(EST, ebuf, eshift, ep, 2 + off + 1);
      numBits = ((widths >> 4) + 1) * 4;
      trail = (widths & 0x0f) * 4;
      glen = 2 + (numBits * __popc(mask) + 7) / 8;
    }
    if (g == my_group) { gOff = off; gBits = numBits; gTrail = trail; gMask = mask; }
    off += glen;
  }
  const int bit = b & 7;
  const uint32_t in_mask = gMask & (1u << bit);
  int slot = __popc(gMask & ((1u << bit) - 1));
  int bitpos = slot * gBits;
  int koff = 2 + gOff + 2 + (bitpos >> 3);
  uint64_t w64 = estream_w64(EST, ebuf, eshift, ep, koff);
  uint32_t b8 = estream_byte(EST, ebuf, eshift, ep, koff + 8);
  int64_t delta = 0;
  if (live && in_mask) {
    int sh = bitpos & 7;
    uint64_t v = w64 >> sh;
    if (gBits > 64 - sh) v |= (uint64_t)b8 << (64 - sh);
    uint64_t m = gBits >= 64 ? ~0ULL : ((1ULL << gBits) - 1);
    delta = (int64_t)((v & m) << gTrail);
  }
  return delta;
}

// uniform-branch dispatcher: big elements (stage window overrun) take the
// unstaged global-load path; everything else stays on the scalarized walk
__device__ __forceinline__ int64_t h2_parse_buf(const uint8_t* ep, int elen,
                                                int nb, int b, bool live,
                                                int lane, uint32_t ebuf,
                                                int eshift) {
  if (elen + 14 + eshift <= 256)
    return h2_parse_est<true>(ep, elen, nb, b, live, lane, ebuf, eshift);
  return h2_parse_est<false>(ep, elen, nb, b, live, lane, ebuf, eshift);
}

__device__ __forceinline__ int64_t h2_parse(const uint8_t* ep, int elen,
                                            int nb, int b, bool live,
                                            int lane) {
  return h2_parse_buf(ep, elen, nb, b, live, lane, estream_stage(ep, lane),
                      (int)((uintptr_t)ep & 3));
}

struct H2Cursor {
  const uint8_t* ep;     // current element start (u16 len prefix)
  const uint8_t* sp;     // NEXT section header
  const uint8_t* sb;     // 4-aligned base of the 256-B register stage (null = none)
  uint32_t sbuf;         // this lane's dword of the stage
  uint32_t sbuf2;        // prefetched next 256-B window (load in flight)
  int elen;              // current element's payload length
  int sect_left;         // elements left in section AFTER the current one
  int c;                 // chunk index within the series
  int e_local, e_global; // element indices (current)
  int nrows;             // rows in current chunk
  bool decoded;          // val_b holds the current element
  bool sect_first;       // current element is its section's base
  double base_b;         // per-lane section base value
  double val_b;          // per-lane raw value of current element (if decoded)
  double C_b;            // per-lane correction total at current position
  double Centry_b;       // C at current chunk entry (after boundary drop)
  double prevlast_b;     // raw value of the previous chunk's last element
  double psum_b;         // per-lane running value prefix (HFUNC==1)
};

// HFUNC 0 = counter-corrected rate (HistRateFunction); 1 = SumOverTime of
// the histograms (raw bucket sums — SumOverTimeChunkedFunctionH, no
// corrections; both cursors then decode every element they pass and carry a
// per-bucket running prefix). out_max/out_min (nullable) add the otel
// companion-column max/min per window, merged across series with
// maxIgnoreNaN/minIgnoreNaN (HistMaxMinSumAggregator).
template <int MINW, int HFUNC>
__global__ __launch_bounds__(H2_WAVES * 64, MINW)
void hist2_kernel(const uint8_t* __restrict__ blob, DirSoA dir,
                  const uint64_t* __restrict__ max_off,
                  const uint64_t* __restrict__ min_off,
                  const int32_t* __restrict__ series_first,
                  const int32_t* __restrict__ series_nchunks,
                  const int32_t* __restrict__ group_ids,
                  int num_series,
                  int64_t qstart, int64_t qstep, int64_t qwindow,
                  int num_windows, int nb,
                  double* __restrict__ out_sums,   // [G × W × nb]
                  double* __restrict__ out_cnt,    // [G × W]
                  double* __restrict__ out_max,    // [G × W] or null
                  double* __restrict__ out_min,    // [G × W] or null
                  int abl) {   // perf ablation: 1=skip emits, 2=skip decodes
  __shared__ int64_t tsS_all[H2_WAVES][H2_ROWS];
  __shared__ int64_t tsE_all[H2_WAVES][H2_ROWS];
  const int wave = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  const int b = lane;
  const bool live = b < nb;
  const bool tmg = (abl & 8) != 0;   // s_memtime phase split into out_cnt[0..3]
  uint64_t tAdv = 0, tDec = 0, tEmit = 0, tOther = 0, tc = 0;

  for (int sid = blockIdx.x * H2_WAVES + wave; sid < num_series;
       sid += gridDim.x * H2_WAVES) {
    const int first = series_first[sid];
    const int nchunks = series_nchunks[sid];
    if (nchunks < 1) continue;
    const int grp = group_ids[sid];

    // per-cursor persistent 256-B register stage: one coalesced global load
    // serves the element hops, section headers and parses of the next ~4-8
    // elements (the raw dependent d_u16 hop loads were the serial chain —
    // ~16k cycles per element of wall, sweep ablate data). Returns the byte
    // offset of ptr within the stage, restaging when [ptr, ptr+need) leaves it.
    // dual-buffer linear streaming: sbuf2 always holds an ISSUED load of the
    // NEXT 256-B window (the stream is consumed forward), so the common
    // restage is a register swap whose vmcnt wait overlapped the previous
    // ~2-3 elements of parse work instead of a cold ~600-cycle stall. A cold
    // (re-anchoring) restage only happens at chunk opens and when an element
    // straddles the window boundary. The +512 blob tail pad covers the
    // prefetch past the last vector.
    auto ensure_stage = [&](H2Cursor& cu, const uint8_t* ptr, int need) -> int {
      int off = (int)(ptr - cu.sb);
      if (cu.sb == nullptr || off < 0 || off + need > 256) {
        if (cu.sb != nullptr && off >= 256 && off + need <= 512) {
          cu.sb += 256;                        // swap in the prefetched window
          cu.sbuf = cu.sbuf2;
        } else {
          // cold anchor at ptr (chunk open, straddling element, or jump)
          cu.sb = (const uint8_t*)((uintptr_t)ptr & ~(uintptr_t)3);
          cu.sbuf = estream_stage(cu.sb, lane);
        }
        cu.sbuf2 = estream_stage(cu.sb + 256, lane);   // issue next prefetch
        off = (int)(ptr - cu.sb);
      }
      return off;
    };
    auto staged_byte = [&](H2Cursor& cu, const uint8_t* ptr) -> uint32_t {
      int off = ensure_stage(cu, ptr, 1);
      return estream_byte_uni(true, cu.sbuf, 0, ptr, off);
    };
    auto staged_u16 = [&](H2Cursor& cu, const uint8_t* ptr) -> uint32_t {
      int off = ensure_stage(cu, ptr, 2);
      return estream_byte_uni(true, cu.sbuf, 0, ptr, off) |
             (estream_byte_uni(true, cu.sbuf, 0, ptr, off + 1) << 8);
    };

    // open a chunk for a cursor: decode its timestamps into the cursor's LDS
    // buffer and position at element 0 (section base, decoded)
    auto open_chunk = [&](H2Cursor& cu, int c, int64_t* tsbuf) {
      const uint8_t* hv = blob + dir.val_off[first + c];
      DVec tv;
      d_vec_open_wide(blob + dir.ts_off[first + c], &tv, nullptr);
      cu.nrows = dir.num_rows[first + c];
      if (cu.nrows > H2_ROWS || cu.nrows > tv.n) cu.nrows = 0;
      d_decode_chunk<false>(tv, cu.nrows, tsbuf, nullptr, lane);
      d_wait_lds();
      __builtin_amdgcn_wave_barrier();
      const int defsz = d_u16(hv + FDB_HIST_OFF_DEFSIZE);
      cu.sp = hv + FDB_HIST_OFF_DEF + defsz;
      cu.c = c;
      cu.e_local = -1;
      cu.sect_left = 0;
      cu.decoded = false;
      cu.sb = nullptr;
    };

    // decode the current element (value = section base + scanned deltas;
    // a section base element's value is the scan of its own deltas)
    auto decode_cur = [&](H2Cursor& cu) {
      if (cu.decoded) return;
      if (abl & 2) { cu.decoded = true; cu.val_b = 0; cu.base_b = 0; return; }
      const int off = ensure_stage(cu, cu.ep, min(cu.elen + 14, 256));
      int64_t delta = h2_parse_buf(cu.ep, cu.elen, nb, b, live, lane,
                                   cu.sbuf, off);
      int64_t scan = wave_incl_scan_i64(live ? delta : 0, lane);
      if (cu.sect_first) { cu.val_b = (double)scan; cu.base_b = cu.val_b; }
      else cu.val_b = cu.base_b + (double)scan;
      cu.decoded = true;
      if (HFUNC == 1) cu.psum_b += cu.val_b;   // value prefix through current
    };

    // step to the next element; returns false when the series is exhausted.
    // tsbuf is the cursor's chunk-timestamp LDS buffer.
    auto step = [&](H2Cursor& cu, int64_t* tsbuf) -> bool {
      if (cu.e_local + 1 >= cu.nrows) {
        if (cu.c + 1 >= nchunks) return false;
        // leaving a chunk: its last element's raw value feeds the boundary
        // drop detection (and TypeDrop at the next chunk's head)
        decode_cur(cu);
        cu.prevlast_b = cu.val_b;
        open_chunk(cu, cu.c + 1, tsbuf);
      }
      bool new_sect = false;
      if (cu.sect_left == 0) {
        // entering a section; TypeDrop adds apply(e-1) (the element we just
        // ensured is decoded below / at the chunk seam above)
        int stype = (int)staged_byte(cu, cu.sp + 3);
        cu.sect_left = (int)staged_byte(cu, cu.sp + 2);
        int slen = (int)staged_u16(cu, cu.sp);
        cu.ep = cu.sp + 4;
        cu.sp += 4 + slen;
        new_sect = true;
        if (stype == 1 && cu.e_local >= 0) cu.C_b += cu.val_b;
      } else {
        cu.ep += 2 + cu.elen;
      }
      cu.elen = (int)staged_u16(cu, cu.ep);
      cu.sect_left--;
      cu.e_local++;
      cu.e_global++;
      cu.sect_first = new_sect;
      cu.decoded = false;
      if (HFUNC == 1) decode_cur(cu);         // sum mode: full prefix
      else if (new_sect) decode_cur(cu);      // section base always decoded
      else if (cu.sect_left == 0 && cu.e_local + 1 < cu.nrows &&
               staged_byte(cu, cu.sp + 3) == 1)
        decode_cur(cu);   // section last feeds the NEXT section's TypeDrop
                          // correction — peek its type byte and decode only
                          // then (drops are rare; the chunk seam decodes its
                          // own last element unconditionally in step())
      if (cu.e_local == 0) {
        // first element of a chunk: boundary drop detection
        // (Histogram.compare top-bucket-down, Histogram.scala:204-214)
        if (cu.c > 0) {
          unsigned long long diff = __ballot(live && cu.val_b != cu.prevlast_b);
          if (diff) {
            int L = 63 - __clzll(diff);
            int lt = __shfl((int)(cu.val_b < cu.prevlast_b), L);
            if (lt) cu.C_b += cu.prevlast_b;
          }
        }
        cu.Centry_b = cu.C_b;
      }
      return true;
    };

    // initialize both cursors at element 0
    H2Cursor S, E;
    memset(&S, 0, sizeof(S)); memset(&E, 0, sizeof(E));
    S.e_global = -1; E.e_global = -1;
    S.C_b = 0; E.C_b = 0;
    S.psum_b = 0; E.psum_b = 0;
    open_chunk(S, 0, tsS_all[wave]);
    if (S.nrows == 0) continue;
    if (!step(S, tsS_all[wave])) continue;
    open_chunk(E, 0, tsE_all[wave]);
    step(E, tsE_all[wave]);
    bool s_more = true, e_more = true;

    for (int w = 0; w < num_windows; w++) {
      if (tmg) tc = __builtin_amdgcn_s_memtime();
      const int64_t wEnd = qstart + (int64_t)w * qstep;
      const int64_t wStart = wEnd - qwindow;
      // E → last element with ts <= wEnd
      for (;;) {
        int64_t nxt;
        if (E.e_local + 1 < E.nrows) nxt = tsE_all[wave][E.e_local + 1];
        else if (E.c + 1 < nchunks) nxt = dir.start_time[first + E.c + 1];
        else break;
        if (nxt > wEnd) break;
        if (!step(E, tsE_all[wave])) { e_more = false; break; }
      }
      // S → first element with ts >= wStart
      while (s_more && tsS_all[wave][S.e_local] < wStart) {
        if (!step(S, tsS_all[wave])) { s_more = false; break; }
      }
      if (tmg) { uint64_t t2m = __builtin_amdgcn_s_memtime(); tAdv += t2m - tc; tc = t2m; }
      (void)e_more;
      if (!s_more) break;                     // no element >= wStart: done
      const int64_t t1 = tsS_all[wave][S.e_local];
      const int64_t t2 = tsE_all[wave][E.e_local];
      if (t1 > wEnd || t2 < wStart) continue; // empty window
      if (t2 < t1) continue;
      const size_t cell = (size_t)grp * num_windows + w;
      // otel companion columns: NaN-ignoring max/min over the window's rows
      // [S..E], read straight from the raw double vectors chunk by chunk
      if (out_max || out_min) {
        double wmax = NAN, wmin = NAN;
        for (int c2 = S.c; c2 <= E.c; c2++) {
          if (max_off[first + c2] == 0) continue;
          DVec xv, nv;
          d_vec_open_wide(blob + max_off[first + c2], &xv, nullptr);
          d_vec_open_wide(blob + min_off[first + c2], &nv, nullptr);
          const int lo = (c2 == S.c) ? S.e_local : 0;
          const int hi = (c2 == E.c) ? E.e_local : dir.num_rows[first + c2] - 1;
          for (int i = lo + lane; i <= hi; i += 64) {
            double mx = d_dv_at(&xv, i);
            double mn = d_dv_at(&nv, i);
            if (!isnan(mx) && (isnan(wmax) || mx > wmax)) wmax = mx;
            if (!isnan(mn) && (isnan(wmin) || mn < wmin)) wmin = mn;
          }
        }
        for (int off = 32; off > 0; off >>= 1) {
          double o = __shfl_down(wmax, off);
          if (!isnan(o) && (isnan(wmax) || o > wmax)) wmax = o;
          o = __shfl_down(wmin, off);
          if (!isnan(o) && (isnan(wmin) || o < wmin)) wmin = o;
        }
        if (lane == 0) {
          if (out_max && !isnan(wmax)) atomic_min_max_f64(&out_max[cell], wmax, false);
          if (out_min && !isnan(wmin)) atomic_min_max_f64(&out_min[cell], wmin, true);
        }
      }
      if (HFUNC == 0) {
        if (!(t2 > t1)) continue;             // highestTime > lowestTime rule
        // paired decode: issue both cursors' stage loads together and let the
        // two (independent) header-parse chains interleave — the per-element
        // parse latency was 19 of the 36 ms on config #4 (sweep ablate=3)
        if (!S.decoded && !E.decoded && !(abl & 2)) {
          const int offS = ensure_stage(S, S.ep, min(S.elen + 14, 256));
          const int offE = ensure_stage(E, E.ep, min(E.elen + 14, 256));
          int64_t dS = h2_parse_buf(S.ep, S.elen, nb, b, live, lane, S.sbuf,
                                    offS);
          int64_t dE = h2_parse_buf(E.ep, E.elen, nb, b, live, lane, E.sbuf,
                                    offE);
          int64_t sS = wave_incl_scan_i64(live ? dS : 0, lane);
          int64_t sE = wave_incl_scan_i64(live ? dE : 0, lane);
          if (S.sect_first) { S.val_b = (double)sS; S.base_b = S.val_b; }
          else S.val_b = S.base_b + (double)sS;
          if (E.sect_first) { E.val_b = (double)sE; E.base_b = E.val_b; }
          else E.val_b = E.base_b + (double)sE;
          S.decoded = true; E.decoded = true;
        }
        decode_cur(S);
        decode_cur(E);
        if (tmg) { uint64_t t2m = __builtin_amdgcn_s_memtime(); tDec += t2m - tc; tc = t2m; }
        const int numSamples = E.e_global - S.e_global + 1;
        if (abl & 1) continue;
        if (live) {
          // corrections relative to the window's first chunk: the reference's
          // per-window CorrectionMeta starts NoCorrection there
          double v1 = S.val_b + (S.C_b - S.Centry_b);
          double v2 = E.val_b + (E.C_b - S.Centry_b);
          double r = d_extrapolated_rate(wStart, wEnd, numSamples,
                                         t1, v1, t2, v2, true, true);
          atomicAdd(&out_sums[cell * nb + b], r);
        }
        if (lane == 0) atomicAdd(&out_cnt[cell], 1.0);
        if (tmg) { uint64_t t2m = __builtin_amdgcn_s_memtime(); tEmit += t2m - tc; tc = t2m; }
      } else {
        // SumOverTime: prefix difference over [S..E] inclusive
        if (live) {
          double sum_b = E.psum_b - S.psum_b + S.val_b;
          atomicAdd(&out_sums[cell * nb + b], sum_b);
        }
        if (lane == 0) atomicAdd(&out_cnt[cell], 1.0);
      }
    }
    d_wait_lds();
    __builtin_amdgcn_wave_barrier();
  }
  if (tmg && lane == 0) {
    atomicAdd(&out_cnt[0], (double)tAdv);
    atomicAdd(&out_cnt[1], (double)tDec);
    atomicAdd(&out_cnt[2], (double)tEmit);
    atomicAdd(&out_cnt[3], (double)tOther);
  }
}

int32_t fdb_launch_hist2(hipStream_t stream, const uint8_t* blob, DirSoA dir,
                         const uint64_t* max_off, const uint64_t* min_off,
                         const int32_t* series_first,
                         const int32_t* series_nchunks,
                         const int32_t* group_ids, int num_series,
                         int64_t qstart, int64_t qstep, int64_t qwindow,
                         int num_windows, int nb, int hfunc,
                         double* out_sums, double* out_cnt,
                         double* out_max, double* out_min) {
  int grid = (num_series + H2_WAVES - 1) / H2_WAVES;
  int cap = 8192;
  if (const char* g = getenv("FDB_HIST_GRID")) cap = atoi(g);
  if (cap > 0 && grid > cap) grid = cap;
  const char* hw = getenv("FDB_HIST_WAVES");   // occupancy experiment knob
  const bool w4 = hw && atoi(hw) == 4;         // 5 waves/SIMD measured best
  const char* ab = getenv("FDB_HIST_ABLATE");  // 1=skip emits, 2=skip decodes
  const int abl = ab ? atoi(ab) : 0;
  #define H2ARGS blob, dir, max_off, min_off, series_first, series_nchunks,       group_ids, num_series, qstart, qstep, qwindow, num_windows, nb,       out_sums, out_cnt, out_max, out_min, abl
  if (hfunc == 1)
    hipLaunchKernelGGL((hist2_kernel<4, 1>), dim3(grid), dim3(H2_WAVES * 64), 0,
                       stream, H2ARGS);
  else if (w4)
    hipLaunchKernelGGL((hist2_kernel<4, 0>), dim3(grid), dim3(H2_WAVES * 64), 0,
                       stream, H2ARGS);
  else
    hipLaunchKernelGGL((hist2_kernel<5, 0>), dim3(grid), dim3(H2_WAVES * 64), 0,
                       stream, H2ARGS);
  #undef H2ARGS
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    fdb_set_error("hist2_kernel launch failed: %s", hipGetErrorString(e));
    return FDB_ERR;
  }
  return FDB_OK;
}
