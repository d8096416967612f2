// MI355X (gfx950) engine: chunk-scan + range-vector kernels behind the C-ABI.
//
// Replaces, as one batched GPU launch per (shard, query), the per-window virtual
// dispatch of the reference execution engine:
//   ChunkedWindowIterator.doNext        query/.../exec/PeriodicSamplesMapper.scala:293-330
//   WindowedChunkIterator               core/.../store/ChunkSetInfo.scala:445-529
//   ChunkedRangeFunction.addChunks      query/.../rangefn/RangeFunction.scala:101-198
//   ChunkedRateFunctionBase/extrapolatedRate  rangefn/RateFunctions.scala:72-111,230-289
//   gauge over-time functions           rangefn/AggrOverTimeFunctions.scala
//   RangeVectorAggregator.fastReduce    query/.../exec/AggrOverRangeVectors.scala:320-377
//
// Execution model: one 64-lane wavefront per series; 256-thread
// blocks = 4 series; chunks decoded into LDS once per series; windows parallel
// across lanes reading LDS (the ~97% window overlap is served on-chip). Counter
// correction is a wave-wide prefix scan; the chunk→chunk carry is sequential in
// time order exactly like the reference's CorrectionMeta carry.
//
// There is NO CPU fallback: engine creation fails without a HIP device.

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstring>
#include <cstdio>
#include <cmath>
#include <vector>
#include <cstdlib>

#include "chunk_format.h"
#include "scan_common.h"
#include "../../include/filodb_amd.h"

void fdb_set_error(const char* fmt, ...);   // chunk_builder.cpp

// scan_fast.hip: single-chunk-per-series fast path
int32_t fdb_launch_fast_scan(hipStream_t stream, const uint8_t* blob, DirSoA dir,
                             const int32_t* series_first, const int32_t* series_nchunks,
                             const int32_t* group_ids, const int32_t* series_by_group,
                             int num_series,
                             int64_t qstart, int64_t qstep, int64_t qwindow,
                             int num_windows, int func_id, int agg_id, int emit_group,
                             double* out, double* out_cnt, double* out_sq,
                             int phase_mask);
bool fdb_fast_scan_supported(int func_id);

// scan_stream.hip: unbounded multi-chunk general path (chunk summaries +
// per-window walk)
int32_t fdb_launch_summaries(hipStream_t stream, const uint8_t* blob, DirSoA dir,
                             int64_t num_chunks, void* sums);
int64_t fdb_chunksum_bytes(int64_t num_chunks);
bool fdb_stream_walk_supported(int func_id);

// window_sample.hip: per-(series,window) sample-buffer functions (FN 16-18)
bool fdb_window_sample_supported(int func_id);
// agg_extra.hip: t-digest quantile + count_values cell presenters
int32_t fdb_launch_quantile_cells(hipStream_t stream, const double* grid,
                                  const int32_t* sbg, const int32_t* goff,
                                  int ng, int nw, double q, double* out);
int32_t fdb_launch_count_values(hipStream_t stream, const double* grid,
                                const int32_t* sbg, const int32_t* goff,
                                int ng, int nw, int k_cap,
                                double* out_vals, double* out_cnts,
                                int32_t* out_n, int32_t* overflow);

int32_t fdb_launch_window_sample(hipStream_t stream, const uint8_t* blob,
                                 DirSoA dir, const int32_t* series_first,
                                 const int32_t* series_nchunks, int num_series,
                                 int64_t qstart, int64_t qstep, int64_t qwindow,
                                 int num_windows, int func_id, double param,
                                 double param2, double* out, int32_t* overflow);

// hist2.hip: two-cursor histogram walk (unbounded chunks / window ratio)
int32_t fdb_launch_hist2(hipStream_t stream, const uint8_t* blob, DirSoA dir,
                         const uint64_t* max_off, const uint64_t* min_off,
                         const int32_t* series_first,
                         const int32_t* series_nchunks,
                         const int32_t* group_ids, int num_series,
                         int64_t qstart, int64_t qstep, int64_t qwindow,
                         int num_windows, int nb, int hfunc,
                         double* out_sums, double* out_cnt,
                         double* out_max, double* out_min);
int32_t fdb_launch_stream_walk(hipStream_t stream, const uint8_t* blob,
                               DirSoA dir, const void* sums,
                               const int32_t* series_first,
                               const int32_t* series_nchunks, int num_series,
                               int64_t qstart, int64_t qstep, int64_t qend,
                               int64_t qwindow, int num_windows, int func_id,
                               double* out);

#define HIP_CHECK(expr) do { hipError_t _e = (expr); if (_e != hipSuccess) { \
  fdb_set_error("%s failed: %s", #expr, hipGetErrorString(_e)); return FDB_ERR; } } while (0)
#define HIP_CHECK_NULL(expr) do { hipError_t _e = (expr); if (_e != hipSuccess) { \
  fdb_set_error("%s failed: %s", #expr, hipGetErrorString(_e)); return nullptr; } } while (0)

// The round-1 LDS-resident scan kernel (two capacity tiers, 1600 rows / 16
// chunks) is gone: single-chunk series run scan_fast.hip, everything else
// runs scan_stream.hip with no capacity caps.
#define FDB_MAX_ROWS_PER_SERIES 400   // per-chunk row cap (reference default)
#define WAVES_PER_BLOCK 4

// floor division helper used by the hist kernel below
// ---------------------------------------------------------------------------
// histogram scan kernel (BASELINE config #4)
//   histogram_quantile(q, sum(rate(hist[w])) by group)
//   wave = one series, lane = one bucket (<=64); elements walked in time order
//   decoding the sect-delta NibblePack streams in lock-step, with a ring of
//   active windows (window/step+2 <= FDB_HIST_RING slots).
//   Semantics: SectDeltaHistogramReader (HistogramVector.scala:628-737),
//   HistogramRateFunctionBase (RateFunctions.scala:330-400),
//   HistSumRowAggregator.scala:20-29.
// ---------------------------------------------------------------------------
#define FDB_HIST_RING 24
#define FDB_HIST_MAX_CHUNKS 4
#define HIST_WAVES 2

struct HistWs {
  int64_t ts[FDB_MAX_ROWS_PER_SERIES];  // current chunk's timestamps
  double  ring_lo[FDB_HIST_RING][64];   // lowestValue per bucket per active window
  int64_t ring_t1[FDB_HIST_RING];       // lowestTime (start element's timestamp)
  int32_t ring_w[FDB_HIST_RING];        // window id the slot holds (-1 none)
  int32_t ring_e[FDB_HIST_RING];        // start element index (series-global)
  int32_t ring_c0[FDB_HIST_RING];       // chunk index the window started in
};

__device__ __forceinline__ int64_t d_fdiv(int64_t a, int64_t b) {
  return a >= 0 ? a / b : -((-a + b - 1) / b);
}

// floor(a / b) via double reciprocal + exact integer fixup — i64 division is
// software-emulated (hundreds of cycles); this is 2 multiplies + a short walk
__device__ __forceinline__ int64_t d_fdiv_fast(int64_t a, int64_t b, double inv_b) {
  int64_t w = (int64_t)floor((double)a * inv_b);
  while ((w + 1) * b <= a) w++;
  while (w * b > a) w--;
  return w;
}

// MAXC = max chunks per series this instantiation handles. The walk streams
// chunks in time order carrying a running per-bucket correction C; per-window
// values subtract C0[c0] (C at entry of the window's FIRST chunk) so results
// equal the reference's per-window CorrectionMeta that starts NoCorrection at
// that chunk (RangeFunction.scala:138-165). Chunk-boundary drop detection is
// Histogram.compare's top-bucket-down lexicographic order (Histogram.scala:
// 204-214) via ballot + highest-differing-lane broadcast.
template <int MAXC>
__global__ __launch_bounds__(HIST_WAVES * 64, 4)
void hist_scan_kernel(const uint8_t* __restrict__ blob, DirSoA dir,
                      const int32_t* __restrict__ series_first,
                      const int32_t* __restrict__ series_nchunks,
                      const int32_t* __restrict__ group_ids,
                      int num_series,
                      int64_t qstart, int64_t qstep, int64_t qend, int64_t qwindow,
                      int num_windows, int nb,
                      double* __restrict__ out_sums,   // [G × W × nb]
                      double* __restrict__ out_cnt,    // [G × W]
                      int dbg)   // FDB_HIST_TIME: phase cycles into out_cnt[0..3]
{
  __shared__ HistWs ws_all[HIST_WAVES];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  HistWs& ws = ws_all[wave];
  uint64_t t_parse = 0, t_scan = 0, t_win = 0, tt = 0, n_elem = 0;

  for (int sid = blockIdx.x * HIST_WAVES + wave; sid < num_series;
       sid += gridDim.x * HIST_WAVES) {
    const int nchunks = series_nchunks[sid];
    if (nchunks < 1 || nchunks > MAXC) continue;       // guarded at exec
    const int first = series_first[sid];

    for (int r = lane; r < FDB_HIST_RING; r += 64) ws.ring_w[r] = -1;

    const int b = lane;                                 // bucket owned by lane
    const bool live = b < nb;
    double raw = 0, C = 0, base = 0;
    double C0[MAXC];            // C at entry of each chunk (boundary drop incl.)
    double prevLast = 0;        // raw apply(len-1) of the previous chunk
    int64_t prev_ts = 0;        // last timestamp of the previous chunk
    int EG0 = 0;                // series-global index of current chunk's row 0
    const int grp = group_ids[sid];
    const double inv_step = 1.0 / (double)qstep;

    for (int c = 0; c < nchunks; c++) {
    const int ci = first + c;
    const int n = dir.num_rows[ci];
    if (n <= 0 || n > FDB_MAX_ROWS_PER_SERIES) break;  // guarded at exec

    DVec tv;
    d_vec_open(blob + dir.ts_off[ci], &tv);
    d_decode_chunk<false>(tv, n, ws.ts, nullptr, lane);
    d_wait_lds();
    __builtin_amdgcn_wave_barrier();

    // first timestamp of the next chunk: closes windows ending in the gap
    int64_t next_first_ts = 0;
    if (c + 1 < nchunks) {
      DVec tnext;
      d_vec_open(blob + dir.ts_off[ci + 1], &tnext);
      next_first_ts = d_lv_at(&tnext, 0);
    }

    const uint8_t* hv = blob + dir.val_off[ci];
    const int defsz = d_u16(hv + FDB_HIST_OFF_DEFSIZE);
    const uint8_t* sp = hv + FDB_HIST_OFF_DEF + defsz;  // first section

    int sect_left = 0;        // elements left in current section
    int sect_first = 0;       // next element is the section's base element
    const uint8_t* ep = sp;   // element cursor

    // pre-stage element 0 (first element of the first section)
    uint32_t ebuf = estream_stage(sp + 4, lane);
    int eshift = (int)((uintptr_t)(sp + 4) & 3);

    for (int e = 0; e < n; e++) {
      if (dbg) tt = __builtin_amdgcn_s_memtime();
      if (sect_left == 0) {   // enter next section
        int stype = sp[3];
        sect_left = sp[2];
        ep = sp + 4;
        sp += 4 + d_u16(sp);  // advance to the section after this one
        sect_first = 1;
        if (stype == 1 && e > 0) C += raw;  // TypeDrop: corr += apply(e-1)
      }
      // elen's two bytes sit at staged offsets 0-1: always in range
      const int elen = (int)estream_u16(true, ebuf, eshift, ep, 0);
      // whole-element staging test, uniform across the wave (the max read
      // offset is koff+8 <= elen+10; +shift must stay inside the 256 B stage)
      const bool est = elen + 14 + eshift <= 256;
      // issue the stage load for the NEXT element before parsing this one —
      // its latency overlaps the whole parse below
      const uint8_t* nxt = sect_left > 1 ? ep + 2 + elen : sp + 4;
      uint32_t nbuf = estream_stage(nxt, lane);
      int nshift = (int)((uintptr_t)nxt & 3);
      // parse the 8-value group headers serially (uniform across lanes)
      int my_group = b >> 3;
      int off = 0, gOff = 0, gBits = 0, gTrail = 0;
      uint32_t gMask = 0;
      for (int g = 0; g * 8 < nb; g++) {
        uint32_t mask = estream_byte(est, ebuf, eshift, ep, 2 + off);
        int numBits = 0, trail = 0, glen;
        if (mask == 0) {
          glen = 1;
        } else {
          int widths = (int)estream_byte(est, ebuf, eshift, ep, 2 + off + 1);
          numBits = ((widths >> 4) + 1) * 4;
          trail = (widths & 0x0f) * 4;
          glen = 2 + (numBits * __popc(mask) + 7) / 8;
        }
        if (g == my_group) { gOff = off; gBits = numBits; gTrail = trail; gMask = mask; }
        off += glen;
      }
      // extract this lane's value from its group. The stream reads run with
      // the FULL wave active (shuffle sources must be live); only the final
      // delta is predicated.
      const int bit = b & 7;
      const uint32_t in_mask = gMask & (1u << bit);
      int slot = __popc(gMask & ((1u << bit) - 1));
      int bitpos = slot * gBits;
      int koff = 2 + gOff + 2 + (bitpos >> 3);     // byte offset from ep
      uint64_t w64 = estream_w64(est, ebuf, eshift, ep, koff);
      uint32_t b8 = estream_byte(est, ebuf, eshift, ep, koff + 8);
      int64_t delta = 0;
      if (live && in_mask) {
        int sh = bitpos & 7;
        uint64_t v = w64 >> sh;
        if (gBits > 64 - sh) v |= (uint64_t)b8 << (64 - sh);
        uint64_t m = gBits >= 64 ? ~0ULL : ((1ULL << gBits) - 1);
        delta = (int64_t)((v & m) << gTrail);
      }
      ebuf = nbuf; eshift = nshift;
      if (dbg) { uint64_t t = __builtin_amdgcn_s_memtime(); t_parse += t - tt; tt = t; }
      // reconstruct this element's cumulative bucket value
      int64_t scan = wave_incl_scan_i64(live ? delta : 0, lane);
      if (sect_first) { raw = (double)scan; base = raw; }
      else            raw = base + (double)scan;
      sect_first = 0;
      sect_left--;
      ep += 2 + elen;
      if (e == 0) {
        if (c > 0) {
          // detectDropAndCorrection (HistogramVector.scala:670-681):
          // firstValue < lastValue in Histogram.compare's top-bucket-down
          // order => correction += lastValue (per bucket)
          unsigned long long diff = __ballot(live && raw != prevLast);
          if (diff) {
            int L = 63 - __clzll(diff);          // highest differing bucket
            int lt = __shfl((int)(raw < prevLast), L);
            if (lt) C += prevLast;
          }
        }
        C0[c] = C;   // window-local values subtract this (C==0 when c==0)
      }
      const double corrected = raw + C;
      if (dbg) { uint64_t t = __builtin_amdgcn_s_memtime(); t_scan += t - tt; tt = t; }

      // window triggers (inversion of the row-range search)
      const int64_t ts_e = ws.ts[e];
      // sentinels kept near the data so the double conversion stays exact;
      // at chunk boundaries the neighbor timestamp comes from the adjacent
      // chunk so windows starting/ending in the inter-chunk gap fire here
      const int64_t ts_prev = e > 0 ? ws.ts[e - 1]
                            : (c > 0 ? prev_ts : ts_e - ((int64_t)1 << 40));
      const int64_t ts_next = e + 1 < n ? ws.ts[e + 1]
                            : (c + 1 < nchunks ? next_first_ts
                                               : ts_e + ((int64_t)1 << 40));
      // windows starting at e: wStart in (ts_prev, ts_e]
      // (clamp in i64 BEFORE narrowing)
      int64_t ws_lo64 = d_fdiv_fast(ts_prev - qstart + qwindow, qstep, inv_step) + 1;
      int64_t ws_hi64 = d_fdiv_fast(ts_e - qstart + qwindow, qstep, inv_step);
      if (ws_lo64 < 0) ws_lo64 = 0;
      if (ws_hi64 > num_windows - 1) ws_hi64 = num_windows - 1;
      int ws_lo = (int)ws_lo64, ws_hi = (int)ws_hi64;
      for (int w = ws_lo; w <= ws_hi; w++) {
        int slot = w % FDB_HIST_RING;
        if (live) ws.ring_lo[slot][b] = corrected;
        if (lane == 0) {
          ws.ring_w[slot] = w; ws.ring_e[slot] = EG0 + e;
          ws.ring_t1[slot] = ts_e; ws.ring_c0[slot] = c;
        }
      }
      // windows ending at e: wEnd in [ts_e, ts_next)
      int64_t we_lo64 = d_fdiv_fast(ts_e - qstart + qstep - 1, qstep, inv_step);
      int64_t we_hi64 = (e + 1 < n || c + 1 < nchunks)
          ? d_fdiv_fast(ts_next - qstart + qstep - 1, qstep, inv_step) - 1
          : num_windows - 1;
      if (we_lo64 < 0) we_lo64 = 0;
      if (we_hi64 > num_windows - 1) we_hi64 = num_windows - 1;
      int we_lo = (int)we_lo64, we_hi = (int)we_hi64;
      if (ws_hi >= we_lo)   // a slot written this element may be read below
        { d_wait_lds(); __builtin_amdgcn_wave_barrier(); }
      for (int w = we_lo; w <= we_hi; w++) {
        int slot = w % FDB_HIST_RING;
        if (ws.ring_w[slot] != w) continue;     // window never started (empty)
        int e0 = ws.ring_e[slot];
        int64_t t1 = ws.ring_t1[slot];
        if (!(ts_e > t1)) continue;             // highestTime > lowestTime rule
        int64_t wEnd = qstart + (int64_t)w * qstep;
        int64_t wStart = wEnd - qwindow;
        int numSamples = (EG0 + e) - e0 + 1;
        if (live) {
          // correction total at entry of the window's first chunk: the
          // reference's per-window meta starts NoCorrection there
          double c0v = C0[0];
          #pragma unroll
          for (int k = 1; k < MAXC; k++)
            if (ws.ring_c0[slot] == k) c0v = C0[k];
          double r = d_extrapolated_rate(wStart, wEnd, numSamples,
                                         t1, ws.ring_lo[slot][b] - c0v,
                                         ts_e, corrected - c0v, true, true);
          atomicAdd(&out_sums[((size_t)grp * num_windows + w) * nb + b], r);
        }
        if (lane == 0)
          atomicAdd(&out_cnt[(size_t)grp * num_windows + w], 1.0);
      }
      if (dbg) { t_win += __builtin_amdgcn_s_memtime() - tt; n_elem++; }
    }                                           // elements
    prevLast = raw;                             // updateCorrection: RAW apply(n-1)
    prev_ts = ws.ts[n - 1];
    EG0 += n;
    d_wait_lds();
    __builtin_amdgcn_wave_barrier();
    }                                           // chunks
  }
  if (dbg && lane == 0) {
    atomicAdd(&out_cnt[0], (double)t_parse);
    atomicAdd(&out_cnt[1], (double)t_scan);
    atomicAdd(&out_cnt[2], (double)t_win);
    atomicAdd(&out_cnt[3], (double)n_elem);
  }
}

// quantile present step (Histogram.quantile, Histogram.scala:63-108; geometric)
__global__ void hist_quantile_kernel(const double* __restrict__ sums,
                                     const double* __restrict__ cnt,
                                     double* __restrict__ out,
                                     size_t cells, int nb, double q,
                                     double first, double mult) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= cells) return;
  if (!(cnt[i] > 0)) { out[i] = NAN; return; }
  const double* v = sums + i * nb;
  double top = v[nb - 1];
  if (q < 0) { out[i] = -INFINITY; return; }
  if (q > 1) { out[i] = INFINITY; return; }
  if (nb < 2 || !(top > 0)) { out[i] = NAN; return; }
  double rank = q * top;
  int bucket = 0;
  while (v[bucket] < rank) bucket++;
  double bucketStart = bucket == 0 ? 0 : first * pow(mult, bucket - 1);
  double bucketEnd = first * pow(mult, bucket);
  if (bucket == nb - 1 && isinf(bucketEnd)) { out[i] = first * pow(mult, nb - 2); return; }
  if (bucket == 0 && first <= 0) { out[i] = first; return; }
  double count = bucket == 0 ? v[0] : v[bucket] - v[bucket - 1];
  rank -= bucket == 0 ? 0 : v[bucket - 1];
  out[i] = bucketStart + (bucketEnd - bucketStart) * (rank / count);
}

// presentation fixup for aggregated grids (NaN where no contributions; mean for avg)
__global__ void agg_present_kernel(double* out, const double* cnt, const double* sq,
                                   size_t n, int agg_id, int partial) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  if (partial) return;                      // partial mode: leave raw sums + counts
  if (cnt[i] <= 0) { out[i] = NAN; return; }
  if (agg_id == AGG_GROUP) out[i] = 1.0;   // GroupRowAggregator.scala:12-31
  else if (agg_id == AGG_AVG) out[i] = out[i] / cnt[i];
  else if (agg_id == AGG_STDDEV || agg_id == AGG_STDVAR) {
    double mean = out[i] / cnt[i];
    double var = sq[i] / cnt[i] - mean * mean;   // StddevRowAggregator.scala:49-52
    out[i] = agg_id == AGG_STDDEV ? sqrt(var) : var;
  }
}

// top/bottom-k presenter (TopBottomKRowAggregator.scala:29-100): one thread
// per (group, window) walks its group's series in ascending id order over the
// [S×W] grid, maintaining a sorted k-list — identical insertion order to the
// oracle's fold, so results (including ties) match exactly.
__global__ void topk_kernel(const double* __restrict__ grid,
                            const int32_t* __restrict__ sbg,
                            const int32_t* __restrict__ goff,
                            int ng, int nw, int k, int top,
                            double* __restrict__ out, double* __restrict__ ids) {
  size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (size_t)ng * nw) return;
  int g = (int)(idx / nw), w = (int)(idx % nw);
  double best[16], bid[16];
  for (int j = 0; j < 16; j++) { best[j] = NAN; bid[j] = -1; }
  for (int i = goff[g]; i < goff[g + 1]; i++) {
    int s = sbg[i];
    double x = grid[(size_t)s * nw + w];
    if (isnan(x)) continue;
    int pos = -1;
    for (int j = 0; j < k; j++)
      if (isnan(best[j]) || (top ? x > best[j] : x < best[j])) { pos = j; break; }
    if (pos >= 0) {
      for (int j = k - 1; j > pos; j--) { best[j] = best[j - 1]; bid[j] = bid[j - 1]; }
      best[pos] = x; bid[pos] = (double)s;
    }
  }
  for (int j = 0; j < k; j++) {
    out[idx * k + j] = best[j];
    if (ids) ids[idx * k + j] = bid[j];
  }
}

// cross-series group reduction from a per-series [S×W] grid. Replaces the
// contended-atomic accumulation path: the scan writes plain per-series window
// results (identical to AGG_NONE) and this kernel folds each group's
// contiguous members from the group-sorted index — one coalesced read of the
// grid instead of 2-3 f64 RMWs per (series, window) on a hot [G×W] grid.
// Semantics: RangeVectorAggregator.fastReduce (AggrOverRangeVectors.scala:
// 320-377) with the RowAggregator map/reduce rules (NaN rows skipped; MIN/MAX
// stay NaN until a value arrives). Raw sums+counts out — agg_present_kernel
// applies the presentation step, as the atomic path did.
template <int VARIANT>
__global__ void group_reduce_kernel(const double* __restrict__ grid,
                                    const int32_t* __restrict__ sbg,
                                    const int32_t* __restrict__ goff,
                                    int ng, int nw, int agg_id,
                                    double* __restrict__ out,
                                    double* __restrict__ cnt,
                                    double* __restrict__ sq) {
  size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (size_t)ng * nw) return;
  int g = (int)(idx / nw), w = (int)(idx % nw);
  // consecutive threads share g and walk consecutive w, so each member's row
  // grid[s*nw + w..] is read coalesced across the wave
  const int i0 = goff[g], i1 = goff[g + 1];
  if (agg_id == AGG_MIN || agg_id == AGG_MAX) {
    const bool is_min = agg_id == AGG_MIN;
    double acc = NAN, c = 0.0;
    for (int i = i0; i < i1; i++) {
      double x = grid[(size_t)sbg[i] * nw + w];
      if (isnan(x)) continue;
      c += 1.0;
      acc = isnan(acc) || (is_min ? x < acc : x > acc) ? x : acc;
    }
    out[idx] = acc;
    cnt[idx] = c;
    if (sq) sq[idx] = 0.0;
    return;
  }
  if (VARIANT == 0) {
    double acc = 0.0, c = 0.0, s2 = 0.0;
    for (int i2 = i0; i2 < i1; i2++) {
      double x = grid[(size_t)sbg[i2] * nw + w];
      if (isnan(x)) continue;
      c += 1.0;
      acc += x;
      s2 += x * x;
    }
    out[idx] = (agg_id == AGG_COUNT || agg_id == AGG_GROUP) ? c : acc;
    cnt[idx] = c;
    if (sq) sq[idx] = s2;
    return;
  }
  // sum-shaped reductions: 4 independent accumulator chains keep 4 loads in
  // flight (a single chain leaves the loop latency-bound far below the HBM
  // rate). Reordering the member sum is fine: the atomic path this replaced
  // already accumulated in nondeterministic order.
  double a0 = 0, a1 = 0, a2 = 0, a3 = 0;
  double c0 = 0, c1 = 0, q0 = 0, q1 = 0;
  int i = i0;
  for (; i + 3 < i1; i += 4) {
    double x0 = grid[(size_t)sbg[i] * nw + w];
    double x1 = grid[(size_t)sbg[i + 1] * nw + w];
    double x2 = grid[(size_t)sbg[i + 2] * nw + w];
    double x3 = grid[(size_t)sbg[i + 3] * nw + w];
    bool n0 = !isnan(x0), n1 = !isnan(x1), n2 = !isnan(x2), n3 = !isnan(x3);
    a0 += n0 ? x0 : 0.0; a1 += n1 ? x1 : 0.0;
    a2 += n2 ? x2 : 0.0; a3 += n3 ? x3 : 0.0;
    c0 += (double)n0 + (double)n2; c1 += (double)n1 + (double)n3;
    q0 += n0 ? x0 * x0 : 0.0; q1 += n1 ? x1 * x1 : 0.0;
    q0 += n2 ? x2 * x2 : 0.0; q1 += n3 ? x3 * x3 : 0.0;
  }
  for (; i < i1; i++) {
    double x = grid[(size_t)sbg[i] * nw + w];
    if (isnan(x)) continue;
    a0 += x; c0 += 1.0; q0 += x * x;
  }
  double c = c0 + c1;
  out[idx] = (agg_id == AGG_COUNT || agg_id == AGG_GROUP)
                 ? c : (a0 + a1) + (a2 + a3);
  cnt[idx] = c;
  if (sq) sq[idx] = q0 + q1;
}

// ---------------------------------------------------------------------------
// host engine
// ---------------------------------------------------------------------------
struct fdb_engine {
  int device;
  hipStream_t stream;
  int32_t* dev_flag;        // sample-kernel overflow flag (device)
};

struct fdb_dataset {
  uint8_t* blob;
  uint64_t *ts_off, *val_off;
  int64_t *start_time, *end_time;
  int32_t *num_rows;
  int32_t *series_first, *series_nchunks, *group_ids;
  int32_t *series_by_group, *group_offsets;   // group-sorted series index (topk)
  int32_t num_series;
  int64_t num_chunks;
  int64_t payload_bytes;    // sum of vector bytes (algorithmic HBM footprint)
  int64_t total_samples;
  int max_group;            // max group id seen (for validation)
  int max_rows;             // max rows in one series (capacity tier selection)
  int max_chunks;           // max chunks in one series
  int max_chunk_rows;       // max rows in one chunk (hist per-chunk LDS cap)
  int has_hist;             // dataset holds sect-delta histogram vectors
  int fast_ok;              // single-chunk series, chunk spans fit i32 ms
  void* sums;               // ChunkSum[num_chunks] for the streaming walk
  uint64_t *max_off, *min_off;  // hist companion column offsets (0 = absent)
  int has_mm;               // every hist chunk carries max/min columns
  int has_sc;               // every scalar chunk carries a count column
};

extern "C" fdb_engine_t* fdb_engine_create(int32_t device) {
  int count = 0;
  hipError_t e = hipGetDeviceCount(&count);
  if (e != hipSuccess || count == 0) {
    fdb_set_error("no HIP device present (hipGetDeviceCount: %s) — "
                  "filodb_amd has no CPU fallback", hipGetErrorString(e));
    return nullptr;
  }
  if (device < 0 || device >= count) { fdb_set_error("bad device %d", device); return nullptr; }
  HIP_CHECK_NULL(hipSetDevice(device));
  auto* eng = new fdb_engine();
  eng->device = device;
  if (hipStreamCreate(&eng->stream) != hipSuccess) {
    fdb_set_error("hipStreamCreate failed");
    delete eng;
    return nullptr;
  }
  if (hipMalloc(&eng->dev_flag, 4) != hipSuccess ||
      hipMemset(eng->dev_flag, 0, 4) != hipSuccess) {
    fdb_set_error("engine flag allocation failed");
    (void)hipStreamDestroy(eng->stream);
    delete eng;
    return nullptr;
  }
  return eng;
}

extern "C" void fdb_engine_destroy(fdb_engine_t* e) {
  if (!e) return;
  (void)hipFree(e->dev_flag);
  (void)hipStreamDestroy(e->stream);
  delete e;
}

hipStream_t fdb_engine_stream(fdb_engine_t* e) { return e->stream; }

extern "C" int32_t fdb_engine_synchronize(fdb_engine_t* e) {
  HIP_CHECK(hipStreamSynchronize(e->stream));
  return FDB_OK;
}

extern "C" void fdb_dataset_destroy(fdb_dataset_t* d) {
  if (!d) return;
  (void)hipFree(d->blob); (void)hipFree(d->ts_off); (void)hipFree(d->val_off);
  (void)hipFree(d->start_time); (void)hipFree(d->end_time); (void)hipFree(d->num_rows);
  (void)hipFree(d->series_first); (void)hipFree(d->series_nchunks); (void)hipFree(d->group_ids);
  (void)hipFree(d->series_by_group); (void)hipFree(d->group_offsets);
  (void)hipFree(d->sums);
  (void)hipFree(d->max_off); (void)hipFree(d->min_off);
  delete d;
}

extern "C" int64_t fdb_dataset_bytes(const fdb_dataset_t* d) { return d->payload_bytes; }
extern "C" int64_t fdb_dataset_samples(const fdb_dataset_t* d) { return d->total_samples; }

extern "C" fdb_dataset_t* fdb_dataset_upload(fdb_engine_t* e, const fdb_store_t* s) {
  fdb_view_t view;
  if (fdb_store_view(s, &view) != FDB_OK) return nullptr;
  HIP_CHECK_NULL(hipSetDevice(e->device));

  const fdb_dir_entry_t* dir = (const fdb_dir_entry_t*)view.dir;
  int64_t nc = view.num_chunks;

  // SoA host staging
  std::vector<uint64_t> ts_off(nc), val_off(nc), mx_off(nc), mn_off(nc);
  std::vector<int64_t> st(nc), en(nc);
  std::vector<int32_t> nr(nc);
  int64_t payload = 0, samples = 0;
  int all_mm = 1, any_mm = 0;
  for (int64_t i = 0; i < nc; i++) {
    ts_off[i] = dir[i].ts_off; val_off[i] = dir[i].val_off;
    mx_off[i] = dir[i].max_off; mn_off[i] = dir[i].min_off;
    if (dir[i].max_off) any_mm = 1; else all_mm = 0;
    st[i] = dir[i].start_time; en[i] = dir[i].end_time; nr[i] = dir[i].num_rows;
    uint32_t tl, vl;
    memcpy(&tl, view.blob + dir[i].ts_off, 4);
    memcpy(&vl, view.blob + dir[i].val_off, 4);
    payload += (int64_t)tl + 4 + (int64_t)vl + 4;
    if (dir[i].max_off) {
      uint32_t xl, nl;
      memcpy(&xl, view.blob + dir[i].max_off, 4);
      memcpy(&nl, view.blob + dir[i].min_off, 4);
      payload += (int64_t)xl + 4 + (int64_t)nl + 4;
    }
    samples += dir[i].num_rows;
  }
  int max_group = 0, max_rows = 0, max_chunks = 0, max_chunk_rows = 0;
  int has_hist = 0, has_scalar = 0, spans_fit_i32 = 1;
  for (int64_t i2 = 0; i2 < nc; i2++) {
    uint16_t wf;
    memcpy(&wf, view.blob + dir[i2].val_off + 4, 2);
    if (wf == FDB_WF_HIST_SECTDELTA) has_hist = 1; else has_scalar = 1;
    if (dir[i2].num_rows > max_chunk_rows) max_chunk_rows = dir[i2].num_rows;
    if (dir[i2].end_time - dir[i2].start_time >= (int64_t)1 << 31) spans_fit_i32 = 0;
  }
  if (has_hist && has_scalar) {
    fdb_set_error("mixed histogram and scalar series in one store are not "
                  "supported — upload them as separate datasets");
    return nullptr;
  }
  for (int32_t sid = 0; sid < view.num_series; sid++) {
    if (view.group_ids[sid] > max_group) max_group = view.group_ids[sid];
    int total = 0;
    for (int c = 0; c < view.series_nchunks[sid]; c++)
      total += dir[view.series_first[sid] + c].num_rows;
    if (total > max_rows) max_rows = total;
    if (view.series_nchunks[sid] > max_chunks) max_chunks = view.series_nchunks[sid];
  }
  if (max_chunk_rows > 400) {
    fdb_set_error("chunk has %d rows; the reference's chunk cap is 400 "
                  "(filodb-defaults.conf:835)", max_chunk_rows);
    return nullptr;
  }

  auto* d = new fdb_dataset();
  memset(d, 0, sizeof(*d));
  d->num_series = view.num_series;
  d->num_chunks = nc;
  d->payload_bytes = payload;
  d->total_samples = samples;
  d->max_group = max_group;
  d->max_rows = max_rows;
  d->max_chunks = max_chunks;
  d->max_chunk_rows = max_chunk_rows;
  d->has_hist = has_hist;
  d->fast_ok = !has_hist && max_chunks <= 1 && max_chunk_rows <= 400 &&
               spans_fit_i32;
  d->has_mm = has_hist && any_mm && all_mm;
  d->has_sc = has_scalar && any_mm && all_mm;   // count column rides max_off

  auto upload = [&](void** dst, const void* src, size_t bytes) -> bool {
    if (hipMalloc(dst, bytes ? bytes : 8) != hipSuccess) return false;
    return hipMemcpy(*dst, src, bytes, hipMemcpyHostToDevice) == hipSuccess;
  };
  // group-sorted series index for the top/bottom-k presenter: stable ascending
  // series order per group (ties then resolve identically to the oracle's fold)
  std::vector<int32_t> goff((size_t)max_group + 2, 0);
  for (int32_t s2 = 0; s2 < view.num_series; s2++) goff[(size_t)view.group_ids[s2] + 1]++;
  for (size_t g = 1; g < goff.size(); g++) goff[g] += goff[g - 1];
  std::vector<int32_t> sbg((size_t)view.num_series);
  {
    std::vector<int32_t> cur(goff.begin(), goff.end() - 1);
    for (int32_t s2 = 0; s2 < view.num_series; s2++)
      sbg[(size_t)cur[(size_t)view.group_ids[s2]]++] = s2;
  }

  // +512 B tail pad: the hist walk's wave-staged element reads may extend up
  // to 256 B past the last element's start, and its linear stage PREFETCH one
  // further 256-B window beyond that
  auto upload_pad = [&](void** dst, const void* src, size_t bytes) -> bool {
    if (hipMalloc(dst, bytes + 512) != hipSuccess) return false;
    if (hipMemset((char*)*dst + bytes, 0, 512) != hipSuccess) return false;
    return hipMemcpy(*dst, src, bytes, hipMemcpyHostToDevice) == hipSuccess;
  };
  bool ok = upload_pad((void**)&d->blob, view.blob, (size_t)view.blob_len)
    && upload((void**)&d->ts_off, ts_off.data(), nc * 8)
    && upload((void**)&d->val_off, val_off.data(), nc * 8)
    && upload((void**)&d->start_time, st.data(), nc * 8)
    && upload((void**)&d->end_time, en.data(), nc * 8)
    && upload((void**)&d->num_rows, nr.data(), nc * 4)
    && upload((void**)&d->series_first, view.series_first, (size_t)view.num_series * 4)
    && upload((void**)&d->series_nchunks, view.series_nchunks, (size_t)view.num_series * 4)
    && upload((void**)&d->group_ids, view.group_ids, (size_t)view.num_series * 4)
    && upload((void**)&d->series_by_group, sbg.data(), sbg.size() * 4)
    && upload((void**)&d->group_offsets, goff.data(), goff.size() * 4)
    && upload((void**)&d->max_off, mx_off.data(), nc * 8)
    && upload((void**)&d->min_off, mn_off.data(), nc * 8);
  if (!ok) {
    fdb_set_error("device upload failed (out of HBM?)");
    fdb_dataset_destroy(d);
    return nullptr;
  }
  // chunk summaries for the streaming walk (query-independent; skipped when
  // every series takes the single-chunk fast path)
  if (has_scalar && !d->fast_ok) {
    if (hipMalloc(&d->sums, (size_t)fdb_chunksum_bytes(nc)) != hipSuccess) {
      fdb_set_error("summary allocation failed (out of HBM?)");
      fdb_dataset_destroy(d);
      return nullptr;
    }
    DirSoA dirx{d->ts_off, d->val_off, d->start_time, d->end_time, d->num_rows};
    if (fdb_launch_summaries(e->stream, d->blob, dirx, nc, d->sums) != FDB_OK) {
      fdb_dataset_destroy(d);
      return nullptr;
    }
  }
  return d;
}

// launch dispatch over the func template parameter
// the fast single-chunk kernel handles this (dataset, query) pair?
static bool fast_eligible(const fdb_dataset_t* d, const fdb_query_t* q) {
  const char* v = getenv("FDB_FAST");             // perf/parity experiments
  if (v && atoi(v) == 0) return false;
  return d->fast_ok && fdb_fast_scan_supported(q->func_id) &&
         q->step > 0 && q->step < ((int64_t)1 << 31) && q->window >= 0;
}

static int32_t launch_scan(fdb_engine_t* e, const fdb_dataset_t* d, const fdb_query_t* q,
                           double* dev_out, double* dev_cnt, double* dev_sq,
                           const uint64_t* val_override = nullptr) {
  DirSoA dir{d->ts_off, val_override ? val_override : d->val_off,
             d->start_time, d->end_time, d->num_rows};
  int nw = fdb_num_windows(q);
  if (fast_eligible(d, q) && q->agg_id == AGG_NONE) {
    return fdb_launch_fast_scan(e->stream, d->blob, dir, d->series_first,
                                d->series_nchunks, d->group_ids,
                                d->series_by_group, d->num_series,
                                q->start, q->step, q->window, nw,
                                q->func_id, AGG_NONE, /*emit_group=*/0,
                                dev_out, dev_cnt, dev_sq, q->_pad);
  }
  // sample-buffer functions (quantile/MAD/predict_linear) have their own
  // per-(series,window) kernel, on any dataset shape
  if (fdb_window_sample_supported(q->func_id)) {
    return fdb_launch_window_sample(e->stream, d->blob, dir, d->series_first,
                                    d->series_nchunks, d->num_series,
                                    q->start, q->step, q->window, nw,
                                    q->func_id, q->param, q->param2,
                                    dev_out, e->dev_flag);
  }
  // everything else: the unbounded summary+walk general path. It writes the
  // per-series [S×W] grid only (aggregation is the two-phase reduce outside).
  if (!fdb_stream_walk_supported(q->func_id)) {
    fdb_set_error("bad func_id %d", q->func_id);
    return FDB_ERR_BADARG;
  }
  if (!d->sums) {
    fdb_set_error("dataset has no chunk summaries (internal)");
    return FDB_ERR;
  }
  (void)dev_cnt; (void)dev_sq;
  return fdb_launch_stream_walk(e->stream, d->blob, dir, d->sums,
                                d->series_first, d->series_nchunks,
                                d->num_series, q->start, q->step, q->end,
                                q->window, nw, q->func_id, dev_out);
}

static int32_t run_query(fdb_engine_t* e, const fdb_dataset_t* d, const fdb_query_t* q,
                         double* out, double* out_counts, int32_t out_on_device,
                         int32_t warmup, int32_t iters, double* avg_ms) {
  HIP_CHECK(hipSetDevice(e->device));
  if (d->has_hist) {
    fdb_set_error("histogram dataset: use fdb_query_exec_hist");
    return FDB_ERR_BADARG;
  }
  int nw = fdb_num_windows(q);
  if (nw <= 0) { fdb_set_error("bad window params"); return FDB_ERR_BADARG; }
  if (q->window > ((int64_t)1 << 33)) {   // d_div1000 exactness domain
    fdb_set_error("window length > 2^33 ms (~99 days) unsupported");
    return FDB_ERR_BADARG;
  }
  if (q->func_id == FDB_FN_HOLT_WINTERS &&
      !(q->param >= 0 && q->param <= 1 && q->param2 >= 0 && q->param2 <= 1)) {
    // parseParameters (AggrOverTimeFunctions.scala:1373-1382)
    fdb_set_error("holt_winters sf/tf must be in [0, 1]");
    return FDB_ERR_BADARG;
  }
  const bool is_topk = q->agg_id == AGG_TOPK || q->agg_id == AGG_BOTTOMK;
  const int kk = is_topk ? (int)q->param : 0;
  if (is_topk && (kk < 1 || kk > 16)) {
    fdb_set_error("topk k=%d out of range 1..16 (q.param)", kk);
    return FDB_ERR_BADARG;
  }
  size_t out_len = (q->agg_id == AGG_NONE) ? (size_t)d->num_series * nw
                 : is_topk ? (size_t)q->num_groups * nw * kk
                           : (size_t)q->num_groups * nw;
  if (q->agg_id != AGG_NONE) {
    if (q->num_groups <= 0 || d->max_group >= q->num_groups) {
      fdb_set_error("num_groups %d inconsistent with dataset max group %d",
                    q->num_groups, d->max_group);
      return FDB_ERR_BADARG;
    }
  }

  const bool needs_sq = q->agg_id == AGG_STDDEV || q->agg_id == AGG_STDVAR;
  int partial = (q->agg_id != AGG_NONE && !is_topk && out_counts != nullptr) ? 1 : 0;
  // stddev/stdvar partials ship (raw sums, raw sumsq) stacked: the caller's
  // `out` is [2 × G × W]; shards merge both halves + counts by addition —
  // algebraically StddevRowAggregator.scala:36-52's reduction
  size_t buf_len = (needs_sq && partial) ? out_len * 2 : out_len;

  double *dev_out = out_on_device ? out : nullptr;
  double *dev_cnt = out_on_device ? out_counts : nullptr;
  bool own_out = false, own_cnt = false;
  if (!dev_out) {
    HIP_CHECK(hipMalloc(&dev_out, buf_len * 8));
    own_out = true;
  }
  if (q->agg_id != AGG_NONE && !dev_cnt) {
    HIP_CHECK(hipMalloc(&dev_cnt, out_len * 8));
    own_cnt = true;
  }

  // every exit (including the HIP_CHECK early returns) releases what this call
  // owns — device buffers and events must not leak on a failed launch
  struct Guard {
    double **out_p, **cnt_p, **sq_p, **grid_p;
    bool *own_out_p, *own_cnt_p, *own_sq;
    hipEvent_t *e0, *e1;
    ~Guard() {
      if (*grid_p) (void)hipFree(*grid_p);
      if (*sq_p && *own_sq) (void)hipFree(*sq_p);
      if (*own_out_p) (void)hipFree(*out_p);
      if (*own_cnt_p) (void)hipFree(*cnt_p);
      if (*e0) (void)hipEventDestroy(*e0);
      if (*e1) (void)hipEventDestroy(*e1);
    }
  };
  double* dev_sq = nullptr;
  bool own_sq = false;
  double* per_grid = nullptr;
  hipEvent_t ev0 = nullptr, ev1 = nullptr;
  Guard guard{&dev_out, &dev_cnt, &dev_sq, &per_grid,
              &own_out, &own_cnt, &own_sq, &ev0, &ev1};
  if (needs_sq) {
    if (partial) dev_sq = dev_out + out_len;   // second half of the caller grid
    else { HIP_CHECK(hipMalloc(&dev_sq, out_len * 8)); own_sq = true; }
  }
  HIP_CHECK(hipEventCreate(&ev0));
  HIP_CHECK(hipEventCreate(&ev1));

  // aggregated queries: the fused-group fast path folds fastReduce into the
  // scan (per-lane register partials + one atomic burst per group change) and
  // never materializes the [S×W] grid; other shapes run two-phase — the scan
  // fills an internal per-series grid with plain stores, then a presenter
  // (topk_kernel / group_reduce_kernel) folds it along the group-sorted index
  const bool is_quant = q->agg_id == AGG_QUANTILE;
  if (is_quant && out_counts) {
    fdb_set_error("quantile aggregation has no partial (multi-shard) mode — "
                  "digest shipping is not implemented");
    return FDB_ERR_BADARG;
  }
  // the fused-group emit (no [S×W] intermediate) measured ~6% SLOWER than
  // the two-phase reduce at current scan cost (3.20 vs 3.02 ms on configs[5];
  // the scan is compute-bound, so skipping the grid re-read does not pay
  // yet) — two-phase is the default, FDB_FUSED_GROUP=1 opts in
  const char* fe = getenv("FDB_FUSED_GROUP");
  const bool fused = (fe && atoi(fe) == 1) && !is_quant &&
                     q->agg_id != AGG_NONE && !is_topk &&
                     nw <= 256 && fast_eligible(d, q);
  fdb_query_t qscan = *q;
  if (q->agg_id != AGG_NONE && !fused) {
    HIP_CHECK(hipMalloc(&per_grid, (size_t)d->num_series * nw * 8));
    qscan.agg_id = AGG_NONE;
  }

  int total_runs = warmup + iters;
  float ms_sum = 0;
  for (int it = 0; it < total_runs; it++) {
    bool timed = it >= warmup;
    if (timed) HIP_CHECK(hipEventRecord(ev0, e->stream));
    if (fused) {
      // atomically-accumulated grids: reset per launch. MIN/MAX start the
      // value grid at NaN (atomic_min_max_f64's empty-cell marker).
      const bool mm = q->agg_id == AGG_MIN || q->agg_id == AGG_MAX;
      HIP_CHECK(hipMemsetAsync(dev_out, mm ? 0xFF : 0, buf_len * 8, e->stream));
      HIP_CHECK(hipMemsetAsync(dev_cnt, 0, out_len * 8, e->stream));
      if (dev_sq && !(needs_sq && partial))
        HIP_CHECK(hipMemsetAsync(dev_sq, 0, out_len * 8, e->stream));
      DirSoA dirf{d->ts_off, d->val_off, d->start_time, d->end_time, d->num_rows};
      int32_t rc = fdb_launch_fast_scan(
          e->stream, d->blob, dirf, d->series_first, d->series_nchunks,
          d->group_ids, d->series_by_group, d->num_series,
          q->start, q->step, q->window, nw, q->func_id, q->agg_id,
          /*emit_group=*/1, dev_out, dev_cnt, dev_sq, q->_pad);
      if (rc != FDB_OK) return rc;
      if (timed) {
        HIP_CHECK(hipEventRecord(ev1, e->stream));
        HIP_CHECK(hipEventSynchronize(ev1));
        float ms;
        HIP_CHECK(hipEventElapsedTime(&ms, ev0, ev1));
        ms_sum += ms;
      }
      continue;
    }
    int32_t rc = launch_scan(e, d, &qscan, per_grid ? per_grid : dev_out,
                             per_grid ? nullptr : dev_cnt, nullptr);
    if (rc != FDB_OK) return rc;
    if (is_topk) {
      size_t cells = (size_t)q->num_groups * nw;
      topk_kernel<<<(unsigned)((cells + 255) / 256), 256, 0, e->stream>>>(
          per_grid, d->series_by_group, d->group_offsets,
          q->num_groups, nw, kk, q->agg_id == AGG_TOPK ? 1 : 0,
          dev_out, dev_cnt);
      HIP_CHECK(hipGetLastError());
    } else if (is_quant) {
      int32_t rcq = fdb_launch_quantile_cells(e->stream, per_grid,
                                              d->series_by_group,
                                              d->group_offsets, q->num_groups,
                                              nw, q->param, dev_out);
      if (rcq != FDB_OK) return rcq;
    } else if (q->agg_id != AGG_NONE) {
      size_t cells = (size_t)q->num_groups * nw;
      static int variant = -1;
      if (variant < 0) {
        const char* v = getenv("FDB_REDUCE_VARIANT");   // perf experiments
        variant = v ? atoi(v) : 1;
      }
      if (variant == 1)
        group_reduce_kernel<1><<<(unsigned)((cells + 255) / 256), 256, 0,
                                 e->stream>>>(
            per_grid, d->series_by_group, d->group_offsets,
            q->num_groups, nw, q->agg_id, dev_out, dev_cnt, dev_sq);
      else
        group_reduce_kernel<0><<<(unsigned)((cells + 255) / 256), 256, 0,
                                 e->stream>>>(
            per_grid, d->series_by_group, d->group_offsets,
            q->num_groups, nw, q->agg_id, dev_out, dev_cnt, dev_sq);
      HIP_CHECK(hipGetLastError());
    }
    if (timed) {
      HIP_CHECK(hipEventRecord(ev1, e->stream));
      HIP_CHECK(hipEventSynchronize(ev1));
      float ms;
      HIP_CHECK(hipEventElapsedTime(&ms, ev0, ev1));
      ms_sum += ms;
    }
  }
  if (q->agg_id != AGG_NONE && !is_topk && !is_quant) {
    agg_present_kernel<<<(unsigned)((out_len + 255) / 256), 256, 0, e->stream>>>(
        dev_out, dev_cnt, dev_sq, out_len, q->agg_id, partial);
    HIP_CHECK(hipGetLastError());
  }
  HIP_CHECK(hipStreamSynchronize(e->stream));
  if (fdb_window_sample_supported(q->func_id)) {
    int32_t flag = 0;
    HIP_CHECK(hipMemcpy(&flag, e->dev_flag, 4, hipMemcpyDeviceToHost));
    if (flag) {
      (void)hipMemset(e->dev_flag, 0, 4);
      fdb_set_error("a window exceeded the %d-sample buffer of the "
                    "quantile/MAD kernel", 1024);
      return FDB_ERR;
    }
  }

  if (!out_on_device) {
    HIP_CHECK(hipMemcpy(out, dev_out, buf_len * 8, hipMemcpyDeviceToHost));
    if (q->agg_id != AGG_NONE && out_counts)
      HIP_CHECK(hipMemcpy(out_counts, dev_cnt, out_len * 8, hipMemcpyDeviceToHost));
  }
  if (avg_ms) *avg_ms = iters > 0 ? (double)ms_sum / iters : 0.0;
  return FDB_OK;   // guard frees per_grid/dev_sq/owned buffers + events
}


static int32_t run_hist(fdb_engine_t* e, const fdb_dataset_t* d,
                        const fdb_query_t* q, int32_t nb,
                        double* out_bucket_sums, double* out_counts,
                        double* out_max, double* out_min,
                        double* out_quantile, int32_t out_on_device) {
  HIP_CHECK(hipSetDevice(e->device));
  int nw = fdb_num_windows(q);
  if (nw <= 0 || q->num_groups <= 0) { fdb_set_error("bad hist query params"); return FDB_ERR_BADARG; }
  if (q->window > ((int64_t)1 << 33)) {   // d_div1000 exactness domain
    fdb_set_error("window length > 2^33 ms (~99 days) unsupported");
    return FDB_ERR_BADARG;
  }
  if (nb < 1 || nb > 64) { fdb_set_error("num_buckets must be 1..64"); return FDB_ERR_BADARG; }
  if (!d->has_hist) {
    fdb_set_error("not a histogram dataset");
    return FDB_ERR_BADARG;
  }
  const int hfunc = (q->func_id == FDB_FN_SUM_OVER_TIME) ? 1 : 0;
  if (q->func_id != FDB_FN_HIST_RATE && q->func_id != FDB_FN_SUM_OVER_TIME) {
    fdb_set_error("histogram queries support FDB_FN_HIST_RATE and "
                  "FDB_FN_SUM_OVER_TIME (got %d)", q->func_id);
    return FDB_ERR_BADARG;
  }
  if ((out_max || out_min) && !d->has_mm) {
    fdb_set_error("dataset has no max/min companion columns "
                  "(fdb_series_append_hist_mm)");
    return FDB_ERR_BADARG;
  }
  if (d->max_chunk_rows > FDB_MAX_ROWS_PER_SERIES) {
    fdb_set_error("histogram chunk has %d rows; per-chunk cap is %d",
                  d->max_chunk_rows, FDB_MAX_ROWS_PER_SERIES);
    return FDB_ERR_BADARG;
  }
  const char* hv1 = getenv("FDB_HIST_V1");   // round-1 kernel for A/B runs
  const bool use_v1 = hv1 && atoi(hv1) == 1;
  if (use_v1 && (d->max_chunks > FDB_HIST_MAX_CHUNKS ||
                 q->window / q->step + 2 > FDB_HIST_RING ||
                 hfunc == 1 || out_max || out_min)) {
    fdb_set_error("FDB_HIST_V1 supports only rate without companions within "
                  "the v1 ring caps");
    return FDB_ERR_BADARG;
  }
  if (d->max_group >= q->num_groups) { fdb_set_error("num_groups too small"); return FDB_ERR_BADARG; }
  size_t cells = (size_t)q->num_groups * nw;

  double *dev_sums = nullptr, *dev_cnt = nullptr, *dev_quant = nullptr;
  double *dev_max = nullptr, *dev_min = nullptr;
  bool own_sums = true, own_cnt = true, own_quant = false, own_mm = false;
  if (out_on_device) {
    dev_sums = out_bucket_sums; own_sums = dev_sums == nullptr;
    dev_cnt = out_counts; own_cnt = dev_cnt == nullptr;
    dev_quant = out_quantile;
    dev_max = out_max; dev_min = out_min;
  }
  if (!dev_sums) HIP_CHECK(hipMalloc(&dev_sums, cells * nb * 8));
  if (!dev_cnt) HIP_CHECK(hipMalloc(&dev_cnt, cells * 8));
  if (out_quantile && !dev_quant) { HIP_CHECK(hipMalloc(&dev_quant, cells * 8)); own_quant = true; }
  if (out_max && !dev_max) {
    HIP_CHECK(hipMalloc(&dev_max, cells * 8));
    HIP_CHECK(hipMalloc(&dev_min, cells * 8));
    own_mm = true;
  }
  HIP_CHECK(hipMemsetAsync(dev_sums, 0, cells * nb * 8, e->stream));
  HIP_CHECK(hipMemsetAsync(dev_cnt, 0, cells * 8, e->stream));
  if (dev_max) {    // NaN start for the maxIgnoreNaN/minIgnoreNaN merges
    HIP_CHECK(hipMemsetAsync(dev_max, 0xFF, cells * 8, e->stream));
    HIP_CHECK(hipMemsetAsync(dev_min, 0xFF, cells * 8, e->stream));
  }

  DirSoA dir{d->ts_off, d->val_off, d->start_time, d->end_time, d->num_rows};
  int grid = (d->num_series + HIST_WAVES - 1) / HIST_WAVES;
  int hcap = 8192;
  if (const char* g = getenv("FDB_HIST_GRID")) hcap = atoi(g);  // perf experiments
  if (hcap > 0 && grid > hcap) grid = hcap;
  int dbg = getenv("FDB_HIST_TIME") ? 1 : 0;
  if (!use_v1) {
    int32_t rc = fdb_launch_hist2(e->stream, d->blob, dir, d->max_off,
                                  d->min_off, d->series_first,
                                  d->series_nchunks, d->group_ids,
                                  d->num_series, q->start, q->step, q->window,
                                  nw, nb, hfunc, dev_sums, dev_cnt,
                                  dev_max, dev_min);
    if (rc != FDB_OK) return rc;
  } else if (d->max_chunks > 1)
    hipLaunchKernelGGL(hist_scan_kernel<FDB_HIST_MAX_CHUNKS>, dim3(grid),
                       dim3(HIST_WAVES * 64), 0, e->stream,
                       d->blob, dir, d->series_first, d->series_nchunks, d->group_ids,
                       d->num_series, q->start, q->step, q->end, q->window, nw, nb,
                       dev_sums, dev_cnt, dbg);
  else
    hipLaunchKernelGGL(hist_scan_kernel<1>, dim3(grid), dim3(HIST_WAVES * 64), 0,
                       e->stream,
                       d->blob, dir, d->series_first, d->series_nchunks, d->group_ids,
                       d->num_series, q->start, q->step, q->end, q->window, nw, nb,
                       dev_sums, dev_cnt, dbg);
  HIP_CHECK(hipGetLastError());

  if (out_quantile) {
    // bucket scheme (first, mult) from the first hist vector in the blob
    uint64_t voff0;
    HIP_CHECK(hipMemcpy(&voff0, d->val_off, 8, hipMemcpyDeviceToHost));
    double fm[2];
    HIP_CHECK(hipMemcpy(fm, d->blob + voff0 + FDB_HIST_OFF_DEF + 2, 16,
                        hipMemcpyDeviceToHost));
    hist_quantile_kernel<<<(unsigned)((cells + 255) / 256), 256, 0, e->stream>>>(
        dev_sums, dev_cnt, dev_quant, cells, nb, q->param, fm[0], fm[1]);
    HIP_CHECK(hipGetLastError());
  }
  HIP_CHECK(hipStreamSynchronize(e->stream));
  if (!out_on_device) {
    if (out_bucket_sums)
      HIP_CHECK(hipMemcpy(out_bucket_sums, dev_sums, cells * nb * 8, hipMemcpyDeviceToHost));
    if (out_counts)
      HIP_CHECK(hipMemcpy(out_counts, dev_cnt, cells * 8, hipMemcpyDeviceToHost));
    if (out_quantile)
      HIP_CHECK(hipMemcpy(out_quantile, dev_quant, cells * 8, hipMemcpyDeviceToHost));
    if (out_max && dev_max)
      HIP_CHECK(hipMemcpy(out_max, dev_max, cells * 8, hipMemcpyDeviceToHost));
    if (out_min && dev_min)
      HIP_CHECK(hipMemcpy(out_min, dev_min, cells * 8, hipMemcpyDeviceToHost));
  }
  if (own_sums) (void)hipFree(dev_sums);
  if (own_cnt) (void)hipFree(dev_cnt);
  if (own_quant) (void)hipFree(dev_quant);
  if (own_mm) { (void)hipFree(dev_max); (void)hipFree(dev_min); }
  return FDB_OK;
}

extern "C" int32_t fdb_query_exec_hist(fdb_engine_t* e, const fdb_dataset_t* d,
                                       const fdb_query_t* q, int32_t nb,
                                       double* out_bucket_sums, double* out_counts,
                                       double* out_quantile, int32_t out_on_device) {
  return run_hist(e, d, q, nb, out_bucket_sums, out_counts, nullptr, nullptr,
                  out_quantile, out_on_device);
}

// histogram query with otel max/min companion outputs [G × W]
// (SumAndMaxOverTimeFuncHD / CumulativeHistRateAndMinMaxFunction,
//  AggrOverTimeFunctions.scala:612-813; HistMaxMinSumAggregator merges)
extern "C" int32_t fdb_query_exec_hist_mm(fdb_engine_t* e, const fdb_dataset_t* d,
                                          const fdb_query_t* q, int32_t nb,
                                          double* out_bucket_sums, double* out_counts,
                                          double* out_max, double* out_min,
                                          double* out_quantile, int32_t out_on_device) {
  return run_hist(e, d, q, nb, out_bucket_sums, out_counts, out_max, out_min,
                  out_quantile, out_on_device);
}

extern "C" int32_t fdb_query_exec(fdb_engine_t* e, const fdb_dataset_t* d, const fdb_query_t* q,
                                  double* out, double* out_counts, int32_t out_on_device) {
  return run_query(e, d, q, out, out_counts, out_on_device, 0, 1, nullptr);
}

__global__ void avg_div_kernel(double* __restrict__ out,
                               const double* __restrict__ a,
                               const double* __restrict__ b, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = a[i] / b[i];
}

// Downsample avg: AvgWithSumAndCountOverTimeFuncD (AggrOverTimeFunctions.
// scala:820-860) — per window, SumOverTime of the sum column divided by
// SumOverTime of the count column (plain IEEE division, each column with the
// NaN-poison sum semantics). The count column rides the dataset's max_off
// slots, so the second pass is the SAME fast scan with the directory's value
// offsets swapped — no new scan code.
extern "C" int32_t fdb_query_exec_avg_sc(fdb_engine_t* e, const fdb_dataset_t* d,
                                         const fdb_query_t* q, double* out,
                                         int32_t out_on_device) {
  HIP_CHECK(hipSetDevice(e->device));
  if (!d->has_sc) {
    fdb_set_error("dataset has no count columns (fdb_series_append_sc)");
    return FDB_ERR_BADARG;
  }
  int nw = fdb_num_windows(q);
  if (nw <= 0) { fdb_set_error("bad window params"); return FDB_ERR_BADARG; }
  if (q->window > ((int64_t)1 << 33)) {
    fdb_set_error("window length > 2^33 ms unsupported");
    return FDB_ERR_BADARG;
  }
  if (q->agg_id != AGG_NONE) {
    fdb_set_error("avg_sc emits the [series x windows] grid only");
    return FDB_ERR_BADARG;
  }
  fdb_query_t q2 = *q;
  q2.func_id = FDB_FN_SUM_OVER_TIME;
  if (!fast_eligible(d, &q2)) {
    fdb_set_error("avg_sc needs a fast-eligible dataset (single-chunk "
                  "series, spans inside i32 ms)");
    return FDB_ERR_BADARG;
  }
  const size_t cells = (size_t)d->num_series * (size_t)nw;
  double *dsum = nullptr, *dcnt = nullptr, *dout = nullptr;
  int32_t rc = FDB_ERR;
  fdb_set_error("avg_sc: device allocation failed");
  if (hipMalloc(&dsum, cells * 8) != hipSuccess) goto done;
  if (hipMalloc(&dcnt, cells * 8) != hipSuccess) goto done;
  if (out_on_device) dout = out;
  else if (hipMalloc(&dout, cells * 8) != hipSuccess) goto done;
  rc = launch_scan(e, d, &q2, dsum, nullptr, nullptr);
  if (rc != FDB_OK) goto done;
  rc = launch_scan(e, d, &q2, dcnt, nullptr, nullptr, d->max_off);
  if (rc != FDB_OK) goto done;
  {
    const int threads = 256;
    const size_t grid = (cells + threads - 1) / threads;
    hipLaunchKernelGGL(avg_div_kernel, dim3((uint32_t)grid), dim3(threads), 0,
                       e->stream, dout, dsum, dcnt, cells);
    if (hipGetLastError() != hipSuccess) {
      fdb_set_error("avg_div_kernel launch failed");
      rc = FDB_ERR;
      goto done;
    }
  }
  if (hipStreamSynchronize(e->stream) != hipSuccess) { rc = FDB_ERR; goto done; }
  if (!out_on_device &&
      hipMemcpy(out, dout, cells * 8, hipMemcpyDeviceToHost) != hipSuccess) {
    fdb_set_error("avg_sc copy-out failed");
    rc = FDB_ERR;
    goto done;
  }
  rc = FDB_OK;
done:
  (void)hipFree(dsum);
  (void)hipFree(dcnt);
  if (!out_on_device) (void)hipFree(dout);
  return rc;
}

// count_values cross-series aggregation (CountValuesRowAggregator.scala):
// per (group, window) the distinct non-NaN values with frequencies, sorted
// ascending; more than k_cap distinct values in any cell is an error (the
// reference throws at its 1000-value limit). Host-memory outputs:
//   out_vals/out_cnts: [num_groups × num_windows × k_cap]
//   out_n:             [num_groups × num_windows]
extern "C" int32_t fdb_query_exec_count_values(
    fdb_engine_t* e, const fdb_dataset_t* d, const fdb_query_t* q,
    int32_t k_cap, double* out_vals, double* out_cnts, int32_t* out_n) {
  HIP_CHECK(hipSetDevice(e->device));
  if (d->has_hist) { fdb_set_error("histogram dataset"); return FDB_ERR_BADARG; }
  int nw = fdb_num_windows(q);
  if (nw <= 0 || q->num_groups <= 0 || d->max_group >= q->num_groups) {
    fdb_set_error("bad count_values query params");
    return FDB_ERR_BADARG;
  }
  size_t cells = (size_t)q->num_groups * nw;
  double *per_grid = nullptr, *dv = nullptr, *dc = nullptr;
  int32_t* dn = nullptr;
  int32_t rc = FDB_ERR;
  fdb_query_t qscan = *q;
  qscan.agg_id = AGG_NONE;
  if (hipMalloc(&per_grid, (size_t)d->num_series * nw * 8) != hipSuccess ||
      hipMalloc(&dv, cells * (size_t)k_cap * 8) != hipSuccess ||
      hipMalloc(&dc, cells * (size_t)k_cap * 8) != hipSuccess ||
      hipMalloc(&dn, cells * 4) != hipSuccess) {
    fdb_set_error("count_values allocation failed");
    goto done;
  }
  rc = launch_scan(e, d, &qscan, per_grid, nullptr, nullptr);
  if (rc != FDB_OK) goto done;
  rc = fdb_launch_count_values(e->stream, per_grid, d->series_by_group,
                               d->group_offsets, q->num_groups, nw, k_cap,
                               dv, dc, dn, e->dev_flag);
  if (rc != FDB_OK) goto done;
  if (hipStreamSynchronize(e->stream) != hipSuccess) { rc = FDB_ERR; goto done; }
  {
    int32_t flag = 0;
    (void)hipMemcpy(&flag, e->dev_flag, 4, hipMemcpyDeviceToHost);
    if (flag) {
      (void)hipMemset(e->dev_flag, 0, 4);
      fdb_set_error("count_values: a cell exceeded %d distinct values "
                    "(the reference throws at its 1000-value limit)", k_cap);
      rc = FDB_ERR;
      goto done;
    }
  }
  if (hipMemcpy(out_vals, dv, cells * (size_t)k_cap * 8,
                hipMemcpyDeviceToHost) != hipSuccess ||
      hipMemcpy(out_cnts, dc, cells * (size_t)k_cap * 8,
                hipMemcpyDeviceToHost) != hipSuccess ||
      hipMemcpy(out_n, dn, cells * 4, hipMemcpyDeviceToHost) != hipSuccess) {
    fdb_set_error("count_values copy-back failed");
    rc = FDB_ERR;
  }
done:
  (void)hipFree(per_grid); (void)hipFree(dv); (void)hipFree(dc); (void)hipFree(dn);
  return rc;
}

extern "C" int32_t fdb_query_bench(fdb_engine_t* e, const fdb_dataset_t* d, const fdb_query_t* q,
                                   double* out, double* out_counts, int32_t out_on_device,
                                   int32_t warmup, int32_t iters, double* avg_kernel_ms) {
  return run_query(e, d, q, out, out_counts, out_on_device, warmup, iters, avg_kernel_ms);
}
