// Fast scan kernel: the single-chunk-per-series shape (BASELINE configs[1-3]).
//
// Ground-up round-2 redesign of the window phase (profiles/README.md round-1
// findings: the old kernel was latency-bound on per-window search + branchy
// f64 epilogue chains at 4 waves/SIMD). What changed:
//
//  * window row-ranges come from ONE inversion scan over the rows instead of
//    per-window searches: row i computes two floor-divisions and writes the
//    window ranges it bounds into sw[]/ew[] (startRow/endRow per window) —
//    O(rows + windows) total, exact for arbitrary sorted timestamps
//    (replaces DeltaDeltaDataReader.binarySearch/ceilingIndex on the device;
//    semantics: WindowedChunkIterator, ChunkSetInfo.scala:467-511)
//  * timestamps live in LDS as i32 offsets from the chunk's first ts (halves
//    the LDS footprint and read traffic; chunk span is checked at upload)
//  * the extrapolatedRate epilogue keeps the oracle's exact operation
//    sequence (its threshold comparisons are discontinuous — see the window
//    phase); AVG/STDDEV use a per-block reciprocal table (continuous, ≤2 ulp)
//  * window results are branch-lean selects; one store per window
//
// Per-window semantics are IDENTICAL to the round-1 kernel's single-chunk
// stage C (parity-green against the oracle): RateFunctions.scala:72-111,
// 230-289; AggrOverTimeFunctions.scala:553-605,924-1028,1082-1224;
// RangeFunction.scala:595-745.
//
// EMIT=1 ("fused group" emit) folds fastReduce into the scan: each lane keeps
// its windows' group-partials in registers while the wave walks a contiguous
// slab of the group-sorted series index, flushing one atomicAdd burst per
// group change — the [S×W] intermediate grid of the two-phase reduce is never
// materialized (AggrOverRangeVectors.scala:320-377 semantics; raw sums +
// counts out, agg_present_kernel applies the presentation step).

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstring>
#include <cmath>

#include "chunk_format.h"
#include "scan_common.h"
#include "../../include/filodb_amd.h"

void fdb_set_error(const char* fmt, ...);   // chunk_builder.cpp

#define FAST_ROWS 400        // reference chunk row cap (filodb-defaults.conf:835)
#define FAST_TILE 256        // windows per tile (sw/ew LDS arrays)
#define FAST_WAVES 4
#define FAST_DROPS 32        // sparse counter-reset table capacity

// kinds (which auxiliary LDS arrays a function family needs)
#define K_RATE    0
#define K_PFX     1
#define K_PFX_SQ  2
#define K_MINMAX  3
#define K_CHANGES 4
#define K_LAST    5

template <int FUNC> struct FKind { static constexpr int v =
    (FUNC <= FN_DELTA) ? K_RATE :
    (FUNC == FN_SUM || FUNC == FN_AVG || FUNC == FN_COUNT ||
     FUNC == FN_RATE_OVER_DELTA) ? K_PFX :
    (FUNC == FN_STDDEV || FUNC == FN_STDVAR) ? K_PFX_SQ :
    (FUNC == FN_MIN || FUNC == FN_MAX) ? K_MINMAX :
    (FUNC == FN_CHANGES) ? K_CHANGES : K_LAST; };

template <int KIND>
struct FWs {                        // per-wave LDS workspace
  int32_t tso[FAST_ROWS];           // ts - ts0 (i32 offsets)
  double  val[FAST_ROWS];           // raw values, or prefix sums (PFX kinds)
  uint16_t cnt[(KIND == K_PFX || KIND == K_PFX_SQ || KIND == K_CHANGES)
                   ? FAST_ROWS : 1];
  double  sq[KIND == K_PFX_SQ ? FAST_ROWS : 1];
  double  grp[KIND == K_MINMAX ? (FAST_ROWS + 7) / 8 : 1];
  int16_t dpos[KIND == K_RATE ? FAST_DROPS : 1];
  double  dcum[KIND == K_RATE ? FAST_DROPS : 1];
  int16_t sw[FAST_TILE];            // per-window startRow (n = none)
  int16_t ew[FAST_TILE];            // per-window endRow (-1 = none)
};

template <int KIND> struct NeedsInv {   // reciprocal table users (AVG/STDDEV;
  // the rate epilogue keeps the oracle's exact divisions — its threshold
  // comparisons are discontinuous and must round identically, see below)
  static constexpr bool v = KIND == K_PFX || KIND == K_PFX_SQ;
};

// in-chunk correction at row i from the sparse drop table (the step function
// CorrectingDoubleVectorReader :325-342 materializes as corrected[])
template <typename WS>
__device__ __forceinline__ double f_corr_at(const WS& ws, int dcount, bool dense,
                                            int n, int i) {
  if (dense) {      // >FAST_DROPS resets in the chunk: serial recompute (rare)
    double corr = 0, last = -1.7976931348623157e308;
    for (int j = 0; j <= i; j++) {
      double x = ws.val[j];
      if (isnan(x)) x = 0;
      if (x < last) corr += last;
      last = x;
    }
    return corr;
  }
  double c = 0;
  for (int j = 0; j < dcount; j++) {
    if (ws.dpos[j] <= i) c = ws.dcum[j]; else break;
  }
  return c;
}

// MINW = min waves/SIMD the register allocator must meet. PFX_SQ is
// LDS-bound at 3 blocks/CU; the group-emit variant carries 12 accumulator
// registers (4 waves); K_RATE (no reciprocal table, LDS fits 6 blocks/CU) is
// built at BOTH 5 (96 VGPR, no spills) and 6 (80 VGPR, ~19 spilled) — the
// launcher picks via FDB_RATE_WAVES, measured on hardware.
template <int FUNC, int EMIT, int MINW, bool TIMED = false, bool PF = true>
__global__ __launch_bounds__(FAST_WAVES * 64, MINW)
void fast_scan_kernel(const uint8_t* __restrict__ blob, DirSoA dir,
                      const int32_t* __restrict__ series_first,
                      const int32_t* __restrict__ series_nchunks,
                      const int32_t* __restrict__ group_ids,
                      const int32_t* __restrict__ sbg,   // EMIT=1: group-sorted order
                      int num_series,
                      int64_t qstart, int64_t qstep, int64_t qwindow,
                      int num_windows, int agg_id,
                      double* __restrict__ out,
                      double* __restrict__ out_cnt,
                      double* __restrict__ out_sq,
                      int pm)    // bit 3: s_memtime phase split into out[]
{
  constexpr int KIND = FKind<FUNC>::v;
  constexpr bool IS_COUNTER = (FUNC == FN_RATE || FUNC == FN_INCREASE);
  __shared__ FWs<KIND> ws_all[FAST_WAVES];
  __shared__ double inv_tab[NeedsInv<KIND>::v ? FAST_ROWS + 2 : 1];

  // readfirstlane pins the wave id (and everything derived from it — series
  // ids, directory offsets, header pointers) as wave-uniform for the
  // compiler: the whole per-series bookkeeping chain then lives in SGPRs
  // with scalar loads instead of occupying VGPRs
  const int wave = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  FWs<KIND>& ws = ws_all[wave];

  if (NeedsInv<KIND>::v) {
    for (int i = threadIdx.x; i < FAST_ROWS + 2; i += FAST_WAVES * 64)
      inv_tab[i] = 1.0 / (double)i;          // [0] = +inf (never a divisor)
    __syncthreads();
  }

  const double inv_step = 1.0 / (double)qstep;
  const double rate_scale = 1000.0 / (double)qwindow;
  const bool qw32 = qwindow < ((int64_t)1 << 31);   // i32 epilogue eligibility
  const double dur_ms = (double)qwindow;
  (void)pm;
  constexpr bool timing = TIMED;        // perf ablation (clobbers out[])
  uint64_t tD = 0, tM = 0, tI = 0, tW = 0, tt = 0;

  // series iteration: grid-stride for the plain grid emit; a contiguous slab
  // of the group-sorted index for the fused-group emit (so one wave sees each
  // group as a run and flushes once per group change)
  int pos0, pos1, pos_step;
  if (EMIT == 1) {
    const int gw = blockIdx.x * FAST_WAVES + wave;
    const int nwaves = gridDim.x * FAST_WAVES;
    const int slab = (num_series + nwaves - 1) / nwaves;
    pos0 = gw * slab;
    pos1 = min(num_series, pos0 + slab);
    pos_step = 1;
  } else {
    pos0 = blockIdx.x * FAST_WAVES + wave;
    pos1 = num_series;
    pos_step = gridDim.x * FAST_WAVES;
  }

  // EMIT=1 per-lane group accumulators (windows lane+64k, k<4; W <= 256)
  double accS[4], accQ[4], accC[4];
  int cur_grp = -1;
  if (EMIT == 1) {
    #pragma unroll
    for (int k = 0; k < 4; k++) { accS[k] = NAN; accQ[k] = 0; accC[k] = 0; }
  }
  auto flush_group = [&](int g) {
    #pragma unroll
    for (int k = 0; k < 4; k++) {
      int w = lane + 64 * k;
      if (w < num_windows && accC[k] > 0) {
        size_t cell = (size_t)g * num_windows + w;
        if (agg_id == AGG_MIN || agg_id == AGG_MAX)
          atomic_min_max_f64(&out[cell], accS[k], agg_id == AGG_MIN);
        else
          atomicAdd(&out[cell],
                    (agg_id == AGG_COUNT || agg_id == AGG_GROUP) ? accC[k] : accS[k]);
        atomicAdd(&out_cnt[cell], accC[k]);
        if (out_sq) atomicAdd(&out_sq[cell], accQ[k]);
      }
      accS[k] = NAN; accQ[k] = 0; accC[k] = 0;
    }
  };

  // two-level software pipeline over the series loop: the dependent chain
  // series_first[sid] → dir offsets → 32-B vector headers is 3 serial HBM
  // round trips per series; prefetching each level a stage early hides them
  // under the previous series' decode/window work (d_wait_lds is lgkm-only,
  // so prefetch loads stay in flight across the phase fences)
  int pf_sid = 0, pf_first = 0, pf_nch = 0, pf_nr = 0;
  uint64_t pf_toff = 0, pf_voff = 0;
  DVec pf_tv, pf_vv;
  int64_t pf_ts0 = 0;
  bool pf_hdr = false;      // pf_tv/pf_vv/pf_ts0 hold the next series' headers
  if (PF && pos0 < pos1) {
    pf_sid = (EMIT == 1) ? sbg[pos0] : pos0;
    pf_first = series_first[pf_sid];
    pf_nch = series_nchunks[pf_sid];
    pf_toff = dir.ts_off[pf_first];
    pf_voff = dir.val_off[pf_first];
    pf_nr = dir.num_rows[pf_first];
  }
  for (int pos = pos0; pos < pos1; pos += pos_step) {
    if (timing) tt = __builtin_amdgcn_s_memtime();
    int sid, nch, nrows;
    uint64_t toff, voff;
    if (PF) {
      sid = pf_sid; nch = pf_nch; nrows = pf_nr;
      toff = pf_toff; voff = pf_voff;
    } else {
      sid = (EMIT == 1) ? sbg[pos] : pos;
      const int first0 = series_first[sid];
      nch = series_nchunks[sid];
      toff = dir.ts_off[first0];
      voff = dir.val_off[first0];
      nrows = dir.num_rows[first0];
    }
    // prefetch stage 1: next series' directory row ids
    const int npos = pos + pos_step;
    int nsid = 0, nfirst = 0, nnch = 0;
    if (PF && npos < pos1) {
      nsid = (EMIT == 1) ? sbg[npos] : npos;
      nfirst = series_first[nsid];
      nnch = series_nchunks[nsid];
    }

    // ---- decode: the single chunk into LDS (i32 ts offsets + f64 values) ---
    int n = 0;
    int64_t ts0 = 0;
    bool dropped = false;
    if (nch >= 1) {
      DVec tv, vv;
      if (pf_hdr) {                    // stage-3 prefetch landed last series
        tv = pf_tv; vv = pf_vv; ts0 = pf_ts0;
      } else {
        d_vec_open_wide(blob + toff, &tv, &ts0);   // one 32-B load each
        d_vec_open_wide(blob + voff, &vv, nullptr);
      }
      n = nrows;
      if (n > FAST_ROWS || n > tv.n) { n = 0; ts0 = 0; }   // guarded at upload
      if (n > 0) {
        d_decode_ts_offsets(tv, n, ts0, ws.tso, lane);
        d_decode_chunk<true>(vv, n, nullptr, ws.val, lane);
        dropped = vv.dropped;
      }
    }
    // prefetch stage 2: next series' chunk offsets (nfirst landed during the
    // decode above; these land during this series' meta/window phases)
    if (PF && npos < pos1) {
      pf_sid = nsid; pf_first = nfirst; pf_nch = nnch;
      pf_toff = dir.ts_off[nfirst];
      pf_voff = dir.val_off[nfirst];
      pf_nr = dir.num_rows[nfirst];
    }
    d_wait_lds();
    __builtin_amdgcn_wave_barrier();
    if (timing) { uint64_t t = __builtin_amdgcn_s_memtime(); tD += t - tt; tt = t; }

    // ---- meta: kind-specific per-row structures ----------------------------
    int dcount = 0;
    bool dense = false;
    if constexpr (KIND == K_RATE) {
      // counter-reset scan → sparse (position, cumulative correction) table
      // (CorrectingDoubleVectorReader :325-342; NaN→0 like the reference)
      if (dropped && n > 0) {
        double carry_corr = 0;
        double carry_x = -1.7976931348623157e308;
        for (int base = 0; base < n; base += 64) {
          int i = base + lane;
          double raw = (i < n) ? ws.val[i] : 0;
          double x = (i < n && !isnan(raw)) ? raw : 0;
          double px = __shfl_up(x, 1);
          if (lane == 0) px = carry_x;
          double ci = (i < n && x < px) ? px : 0;
          double scan = wave_incl_scan(ci, lane);
          uint64_t mask = __ballot(ci != 0);
          int here = __popcll(mask);
          if (here) {
            if (dcount + here > FAST_DROPS) {
              dense = true;
            } else if (ci != 0) {
              int slot = dcount + __popcll(mask & ((1ULL << lane) - 1));
              ws.dpos[slot] = (int16_t)i;
              ws.dcum[slot] = carry_corr + scan;
            }
            if (!dense) dcount += here;
          }
          carry_corr += __shfl(scan, 63);
          carry_x = __shfl(x, 63);
        }
        if (dense) dcount = 0;
      }
    }
    if constexpr (KIND == K_PFX || KIND == K_PFX_SQ) {
      double carry = 0, carry_sq = 0;
      int ccarry = 0;
      for (int base = 0; base < n; base += 64) {
        int i = base + lane;
        double raw = (i < n) ? ws.val[i] : NAN;
        bool ok = (i < n) && !isnan(raw);
        double x = ok ? raw : 0;
        double s = wave_incl_scan(x, lane);
        int cs = wave_incl_scan_i(ok ? 1 : 0, lane);
        double sqs = 0;
        if constexpr (KIND == K_PFX_SQ) sqs = wave_incl_scan(x * x, lane);
        __builtin_amdgcn_wave_barrier();        // all raw reads precede writes
        if (i < n) {
          ws.val[i] = carry + s;                // val[] becomes the prefix
          ws.cnt[i] = (uint16_t)(ccarry + cs);
          if constexpr (KIND == K_PFX_SQ) ws.sq[i] = carry_sq + sqs;
        }
        carry += __shfl(s, 63);
        ccarry += __shfl(cs, 63);
        if constexpr (KIND == K_PFX_SQ) carry_sq += __shfl(sqs, 63);
      }
    }
    if constexpr (KIND == K_MINMAX) {
      constexpr bool IS_MIN = (FUNC == FN_MIN);
      for (int base = 0; base < n; base += 64) {
        int i = base + lane;
        double x = (i < n) ? ws.val[i] : NAN;
        #pragma unroll
        for (int off = 1; off < 8; off <<= 1) {
          double o = __shfl_xor(x, off);
          if (!isnan(o) && (isnan(x) || (IS_MIN ? o < x : o > x))) x = o;
        }
        if ((lane & 7) == 0 && i < n) ws.grp[i >> 3] = x;
      }
    }
    if constexpr (KIND == K_CHANGES) {
      int ccarry = 0;
      double carry_x = NAN;
      for (int base = 0; base < n; base += 64) {
        int i = base + lane;
        double x = (i < n) ? ws.val[i] : NAN;
        double px = __shfl_up(x, 1);
        if (lane == 0) px = carry_x;
        int ind = (i > 0 && i < n && !isnan(x) && !isnan(px) && x != px) ? 1 : 0;
        int s = wave_incl_scan_i(ind, lane);
        if (i < n) ws.cnt[i] = (uint16_t)(ccarry + s);
        ccarry += __shfl(s, 63);
        carry_x = __shfl(x, 63);
      }
    }
    d_wait_lds();
    __builtin_amdgcn_wave_barrier();
    if (timing) { uint64_t t = __builtin_amdgcn_s_memtime(); tM += t - tt; tt = t; }

    // window time base: wEnd(w) = qstart + w*qstep; offsets are vs ts0
    const int64_t Ae = ts0 - qstart;            // wEnd >= ts  ⟺ w*qstep >= o+Ae
    if (EMIT == 1) {
      const int grp_id = group_ids[sid];
      if (grp_id != cur_grp) {
        if (cur_grp >= 0) flush_group(cur_grp);
        cur_grp = grp_id;
      }
    }

    for (int tb = 0; tb < num_windows; tb += FAST_TILE) {
      const int tn = min(FAST_TILE, num_windows - tb);
      // prefill: no-start sentinel n, no-end sentinel -1
      for (int i = lane; i < tn; i += 64) { ws.sw[i] = (int16_t)n; ws.ew[i] = -1; }
      d_wait_lds();
      __builtin_amdgcn_wave_barrier();

      // ---- inversion scan: rows → window boundaries ------------------------
      // c_i = first w with wEnd >= ts_i   = ceil((o_i + Ae) / qstep)
      // d_i = last  w with wStart <= ts_i = floor((o_i + Ae + qwindow) / qstep)
      // row i ends   windows [c_{i-1}, c_i)   with e = i-1   (wEnd < ts_i)
      // row i starts windows (d_{i-1}, d_i]   with s = i     (wStart > ts_{i-1})
      // (a shuffle-free variant recomputing the neighbor boundaries from
      // ts[i-1] — 4 fdivs/row, no carries — measured SLOWER: the inversion
      // phase grew 8.2 → 11.1 Gcyc; the shfl form stays)
      // when qstep | qwindow (the usual Prometheus shape) one division gives
      // both boundaries: f = floor((o+Ae)/qstep) with its exact mod-zero bit
      // ⇒ ci = f + (mod != 0) [= ceil], di = f + qwindow/qstep. The wider low
      // clamp keeps di's derivation exact down to di = -1 (see fdiv_floor_rem).
      const bool kdiv = (qwindow % qstep) == 0;
      const int Kwin = kdiv ? (int)(qwindow / qstep) : 0;
      int c_carry = 0, d_carry = -1;
      for (int base = 0; base < n; base += 64) {
        const int i = base + lane;
        const bool live = i < n;
        int64_t o = live ? (int64_t)ws.tso[i] : 0;
        int ci, di;
        if (kdiv) {             // wave-uniform branch
          bool nz = false;
          int f = live ? fdiv_floor_rem(o + Ae, qstep, inv_step, num_windows,
                                        -1 - Kwin, &nz) : 0;
          ci = f + (nz ? 1 : 0);
          di = f + Kwin;
          if (!live) { ci = 0; di = 0; }
        } else {
          ci = live ? fdiv_floor_win(o + Ae + qstep - 1, qstep, inv_step,
                                     num_windows) : 0;
          di = live ? fdiv_floor_win(o + Ae + qwindow, qstep, inv_step,
                                     num_windows) : 0;
        }
        int cprev = __shfl_up(ci, 1);
        int dprev = __shfl_up(di, 1);
        if (lane == 0) { cprev = c_carry; dprev = d_carry; }
        if (live) {
          if (i > 0) {
            int lo = max(cprev, tb), hi = min(ci, tb + tn);
            for (int w = lo; w < hi; w++) ws.ew[w - tb] = (int16_t)(i - 1);
          }
          int lo = (i == 0) ? tb : max(dprev + 1, tb);
          int hi = min(di, tb + tn - 1);
          for (int w = lo; w <= hi; w++) ws.sw[w - tb] = (int16_t)i;
        }
        const int lastl = min(63, n - 1 - base);
        c_carry = __shfl(ci, lastl);
        d_carry = __shfl(di, lastl);
      }
      if (n > 0) {          // tail: windows with wEnd >= ts_{n-1} end at n-1
        for (int w = max(c_carry, tb) + lane; w < tb + tn; w += 64)
          ws.ew[w - tb] = (int16_t)(n - 1);
      }
      d_wait_lds();
      __builtin_amdgcn_wave_barrier();
      if (timing) { uint64_t t = __builtin_amdgcn_s_memtime(); tI += t - tt; tt = t; }

      // ---- window phase: 4 windows per lane --------------------------------
      // results land through emit_res: plain grid store or fastReduce
      // register-accumulate (NaN rows skipped, RowAggregator semantics)
      auto emit_res = [&](int k, int w, bool wok, double res) {
        if (EMIT == 0) {
          // timing mode suppresses result stores so the per-wave cycle
          // records at out[0..nwaves*4) survive (perf ablation only)
          if (wok && w < num_windows && !timing)
            out[(size_t)sid * num_windows + w] = res;
        } else if (!isnan(res)) {
          switch (agg_id) {
            case AGG_MIN:
              accS[k] = (isnan(accS[k]) || res < accS[k]) ? res : accS[k];
              accC[k] += 1.0; break;
            case AGG_MAX:
              accS[k] = (isnan(accS[k]) || res > accS[k]) ? res : accS[k];
              accC[k] += 1.0; break;
            case AGG_STDDEV: case AGG_STDVAR:
              accQ[k] += res * res;
              accS[k] = isnan(accS[k]) ? res : accS[k] + res;
              accC[k] += 1.0; break;
            case AGG_COUNT: case AGG_GROUP:
              accC[k] += 1.0; break;
            default:   // AGG_SUM / AGG_AVG
              accS[k] = isnan(accS[k]) ? res : accS[k] + res;
              accC[k] += 1.0; break;
          }
        }
      };

      if constexpr (KIND == K_RATE) {
        // ChunkedRateFunctionBase + extrapolatedRate; e>s implies s valid,
        // covers empty and single-sample windows (highestTime>lowestTime).
        // The epilogue keeps the oracle's EXACT operation sequence
        // (RateFunctions.scala:72-111): its durationToZero/threshold
        // comparisons are discontinuous — counters make exact rational ties
        // like v1/delta == 1.1/(numSamples-1) common, and the branch taken
        // then depends on the reference's own FP rounding. All inputs are
        // ts0-relative; extrapolatedRate uses differences only, so
        // offset-domain i64s give bit-identical results.
        // (an explicitly staged variant — all 4 windows' loads before any
        // math — measured SLOWER: 2.84 -> 3.26 ms, register pressure)
        #pragma unroll
        for (int k = 0; k < 4; k++) {
          const int wi = lane + 64 * k;
          const int w = tb + wi;
          const bool wok = wi < tn;
          const int s = wok ? ws.sw[wi] : 1;
          const int e = wok ? ws.ew[wi] : -1;
          double res = NAN;
          if (e > s) {
            const int t1 = ws.tso[s], t2 = ws.tso[e];
            if (t2 > t1) {
              double v1 = ws.val[s], v2 = ws.val[e];
              if (IS_COUNTER && dropped) {
                if (isnan(v1)) v1 = 0;
                if (isnan(v2)) v2 = 0;
                v1 += f_corr_at(ws, dcount, dense, n, s);
                v2 += f_corr_at(ws, dcount, dense, n, e);
              }
              const int64_t wEndOff = (int64_t)w * qstep - Ae;   // wEnd - ts0
              if (qw32)      // uniform: window fits i32 ⇒ so do all three gaps
                res = d_extrapolated_rate_i32(
                    (int32_t)(t1 - (wEndOff - qwindow)), (int32_t)(wEndOff - t2),
                    e - s + 1, t2 - t1, dur_ms, v1, v2,
                    IS_COUNTER, FUNC == FN_RATE);
              else
                res = d_extrapolated_rate(wEndOff - qwindow, wEndOff, e - s + 1,
                                          t1, v1, t2, v2,
                                          IS_COUNTER, FUNC == FN_RATE);
            }
          }
          emit_res(k, w, wok, res);
        }
      } else if constexpr (KIND == K_PFX) {
        #pragma unroll
        for (int k = 0; k < 4; k++) {
          const int wi = lane + 64 * k;
          const bool wok = wi < tn;
          const int s = wok ? ws.sw[wi] : 1;
          const int e = wok ? ws.ew[wi] : -1;
          double res = NAN;
          if (e >= s && e >= 0) {
            double ps = ws.val[e] - (s ? ws.val[s - 1] : 0.0);
            int pc = (int)ws.cnt[e] - (s ? (int)ws.cnt[s - 1] : 0);
            if constexpr (FUNC == FN_SUM) res = pc > 0 ? ps : NAN;
            else if constexpr (FUNC == FN_COUNT) res = (double)pc;
            else if constexpr (FUNC == FN_AVG)
              res = pc > 0 ? ps * inv_tab[pc] : NAN;
            else   // FN_RATE_OVER_DELTA (RateFunctions.scala:424-445)
              res = (pc > 0 ? ps : NAN) * rate_scale;
          }
          emit_res(k, tb + wi, wok, res);
        }
      } else {
      #pragma unroll
      for (int k = 0; k < 4; k++) {
        const int wi = lane + 64 * k;
        const int w = tb + wi;
        const bool wok = wi < tn;
        const int s = wok ? ws.sw[wi] : 0;
        const int e = wok ? ws.ew[wi] : -1;
        double res = NAN;

        if constexpr (KIND == K_PFX_SQ) {
          if (wok && e >= s && e >= 0) {
            double ps = ws.val[e] - (s ? ws.val[s - 1] : 0.0);
            int pc = (int)ws.cnt[e] - (s ? (int)ws.cnt[s - 1] : 0);
            if (pc > 0) {
              double qs = ws.sq[e] - (s ? ws.sq[s - 1] : 0.0);
              double inv = inv_tab[pc];
              double avg = ps * inv;
              double r = qs * inv - avg * avg;
              res = (FUNC == FN_STDDEV) ? sqrt(r) : r;
            }
          }
        } else if constexpr (KIND == K_MINMAX) {
          if (wok && e >= s && e >= 0) {
            constexpr bool IS_MIN = (FUNC == FN_MIN);
            double mm = NAN;
            auto acc = [&](double x) {
              if (!isnan(x) && (isnan(mm) || (IS_MIN ? x < mm : x > mm))) mm = x;
            };
            int ga = (s + 7) >> 3, gb = (e + 1) >> 3;
            if (ga < gb) {
              for (int i = s; i < ga * 8; i++) acc(ws.val[i]);
              for (int g = ga; g < gb; g++) acc(ws.grp[g]);
              for (int i = gb * 8; i <= e; i++) acc(ws.val[i]);
            } else {
              for (int i = s; i <= e; i++) acc(ws.val[i]);
            }
            res = mm;
          }
        } else if constexpr (KIND == K_CHANGES) {
          // single chunk: change-indicator prefix over (s, e]; prev starts NaN
          if (wok && e >= s && e >= 0)
            res = (double)((int)ws.cnt[e] - (int)ws.cnt[s]);
        } else {  // K_LAST family
          if constexpr (FUNC == FN_TIMESTAMP) {
            // TimestampChunkedFunction: no window-start bound; seconds
            if (wok && e >= 0) res = (double)(ts0 + ws.tso[e]) / 1000.0;
          } else if constexpr (FUNC == FN_PRESENT) {
            if (wok && e >= s && e >= 0) {
              double v = ws.val[e];
              if (!isnan(v)) res = 1.0;
              else if (e > 0) res = isnan(ws.val[e - 1]) ? NAN : 1.0;
            }
          } else if constexpr (FUNC == FN_LAST) {
            // ts[e] >= ts[s] >= wStart holds whenever the range is nonempty
            if (wok && e >= s && e >= 0) res = ws.val[e];
          } else {  // FN_ZSCORE (AggrOverTimeFunctions.scala:1592-1603)
            if (wok && e >= s && e >= 0) {
              double sm = NAN, sq = NAN, lastv = NAN;
              int pc = 0;
              for (int i = s; i <= e; i++) {
                double x = ws.val[i];
                if (isnan(x)) continue;
                if (isnan(sm)) { sm = 0; sq = 0; }
                if (i == e) lastv = x;
                sm += x; sq += x * x; pc++;
              }
              if (pc > 0) {
                double avg = sm / pc;
                double sd = sqrt(sq / pc - avg * avg);
                res = (lastv - avg) / sd;
              } else res = isnan(sm) ? sm : 0;
            }
          }
        }

        emit_res(k, w, wok, res);
      }
      }  // generic-kind loop
      d_wait_lds();
      __builtin_amdgcn_wave_barrier();
      if (timing) { uint64_t t = __builtin_amdgcn_s_memtime(); tW += t - tt; tt = t; }
    }  // tiles
    // prefetch stage 3: next series' 32-B vector headers (the stage-2 offset
    // loads landed during the window phase; these fly across the loop edge so
    // the next decode starts with only the payload round-trip outstanding)
    pf_hdr = PF && npos < pos1 && pf_nch >= 1;
    if (pf_hdr) {
      d_vec_open_wide(blob + pf_toff, &pf_tv, &pf_ts0);
      d_vec_open_wide(blob + pf_voff, &pf_vv, nullptr);
    }
  }  // series
  if (EMIT == 1 && cur_grp >= 0) flush_group(cur_grp);
  if (timing && EMIT == 0 && lane == 0) {
    size_t gw = (size_t)blockIdx.x * FAST_WAVES + wave;
    out[gw * 4 + 0] = (double)tD;
    out[gw * 4 + 1] = (double)tM;
    out[gw * 4 + 2] = (double)tI;
    out[gw * 4 + 3] = (double)tW;
  }
}

// ---------------------------------------------------------------------------
// host-side launcher (called from engine.hip's launch dispatch)
// ---------------------------------------------------------------------------
bool fdb_fast_scan_supported(int func_id) {
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

int32_t fdb_launch_fast_scan(hipStream_t stream, const uint8_t* blob, DirSoA dir,
                             const int32_t* series_first, const int32_t* series_nchunks,
                             const int32_t* group_ids, const int32_t* series_by_group,
                             int num_series,
                             int64_t qstart, int64_t qstep, int64_t qwindow,
                             int num_windows, int func_id, int agg_id, int emit_group,
                             double* out, double* out_cnt, double* out_sq,
                             int phase_mask) {
  (void)phase_mask;
  int grid = (num_series + FAST_WAVES - 1) / FAST_WAVES;
  int cap = 16384;      // measured best of {1536, 3072, 8192, 16384}
  if (const char* g = getenv("FDB_GRID")) cap = atoi(g);   // perf experiments
  if (cap > 0 && grid > cap) grid = cap;
  const char* rw = getenv("FDB_RATE_WAVES");   // occupancy experiment knob
  const int rate_w = rw ? atoi(rw) : 6;   // 6-wave prefetch-free measured best (2.48 vs 2.62 ms)
  #define FARGS blob, dir, series_first, series_nchunks, group_ids, \
      series_by_group, num_series, qstart, qstep, qwindow, num_windows, \
      agg_id, out, out_cnt, out_sq, phase_mask
  #define LAUNCH(F, E, W) hipLaunchKernelGGL((fast_scan_kernel<F, E, W>), \
      dim3(grid), dim3(FAST_WAVES * 64), 0, stream, FARGS)
  #define FCASE(F, W) case F: \
    if (emit_group) LAUNCH(F, 1, 4); else LAUNCH(F, 0, W); break
  const bool timed = (phase_mask & 8) != 0;   // s_memtime phase ablation
  switch (func_id) {
    case FN_RATE:
      if (emit_group) LAUNCH(FN_RATE, 1, 4);
      else if (timed) hipLaunchKernelGGL((fast_scan_kernel<FN_RATE, 0, 5, true>),
                                         dim3(grid), dim3(FAST_WAVES * 64), 0,
                                         stream, FARGS);
      else if (rate_w == 6)   // 6 waves/SIMD: prefetch stripped (register diet)
        hipLaunchKernelGGL((fast_scan_kernel<FN_RATE, 0, 6, false, false>),
                           dim3(grid), dim3(FAST_WAVES * 64), 0, stream, FARGS);
      else LAUNCH(FN_RATE, 0, 5);
      break;
    case FN_AVG:
      if (emit_group) LAUNCH(FN_AVG, 1, 4);
      else if (timed) hipLaunchKernelGGL((fast_scan_kernel<FN_AVG, 0, 5, true>),
                                         dim3(grid), dim3(FAST_WAVES * 64), 0,
                                         stream, FARGS);
      else LAUNCH(FN_AVG, 0, 5);
      break;
    FCASE(FN_INCREASE, 6); FCASE(FN_DELTA, 6); FCASE(FN_SUM, 5);
    FCASE(FN_COUNT, 5); FCASE(FN_RATE_OVER_DELTA, 5);
    FCASE(FN_MIN, 5); FCASE(FN_MAX, 5); FCASE(FN_STDDEV, 3); FCASE(FN_STDVAR, 3);
    FCASE(FN_CHANGES, 5); FCASE(FN_LAST, 5); FCASE(FN_PRESENT, 5);
    FCASE(FN_TIMESTAMP, 5); FCASE(FN_ZSCORE, 5);
    default:
      fdb_set_error("fast scan: unsupported func_id %d", func_id);
      return FDB_ERR_BADARG;
  }
  #undef FCASE
  #undef LAUNCH
  #undef FARGS
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    fdb_set_error("fast_scan_kernel launch failed: %s", hipGetErrorString(e));
    return FDB_ERR;
  }
  return FDB_OK;
}
