/* t-digest (merging variant) — restatement of the published algorithm from
 * "Computing Extremely Accurate Quantiles Using t-Digests" (Dunning & Ertl),
 * the data structure behind the reference's cross-series quantile()
 * aggregator (QuantileRowAggregator.scala:5,26 — com.tdunning:t-digest,
 * TDigest.createArrayDigest(100)). The library is a third-party dependency
 * absent from the reference tree, so parity is anchored on
 * the reference's OWN call sites and AggrOverRangeVectorsSpec literals
 * (:132-143, :296-304, :341-352): singleton-centroid interpolation gives
 * quantile(0.5, {2.1, 4.6}) = 3.35 and quantile(0.5, {4.4, 5.4, 5.6}) = 5.4,
 * which this implementation reproduces exactly; larger digests are
 * approximate by design on both sides.
 *
 * This header is compiled UNCHANGED into the CPU oracle (gcc, C99) and the
 * HIP engine — identical operation order, so the engine's results match the
 * oracle's bit-for-bit when values are inserted in the same order (ascending
 * series id; NaNs skipped per reduceMappedRow).
 */
#ifndef FDB_TDIGEST_IMPL_H
#define FDB_TDIGEST_IMPL_H

#include <math.h>
#include <string.h>

#define TD_COMP 100.0
#define TD_NC   220      /* centroid capacity (> 2*compression) */
#define TD_BUF  256      /* unmerged insertion buffer */

#ifdef __HIPCC__
#define TD_FN __device__ __host__ static inline
#else
#define TD_FN static inline
#endif

typedef struct {
  int nc, nbuf;
  double total;
  double tmin, tmax;
  double mean[TD_NC], w[TD_NC];
  double buf[TD_BUF];
} tdigest_t;

TD_FN void td_init(tdigest_t* t) {
  t->nc = 0; t->nbuf = 0; t->total = 0;
  t->tmin = INFINITY; t->tmax = -INFINITY;
}

/* scale function k1: k(q) = COMP * (0.5 + asin(2q-1)/pi); cluster boundary
 * rule: a centroid may absorb the next point while k(q_right)-k(q_left) <= 1 */
TD_FN double td_k(double q) {
  if (q < 0) q = 0;
  if (q > 1) q = 1;
  return TD_COMP * (0.5 + asin(2.0 * q - 1.0) / M_PI);
}
TD_FN double td_q(double k) {
  return 0.5 * (sin(M_PI * (k / TD_COMP - 0.5)) + 1.0);
}

TD_FN void td_flush(tdigest_t* t) {
  if (t->nbuf == 0) return;
  /* sort the buffer ascending (insertion sort: both sides identical and the
   * buffer is small) */
  for (int i = 1; i < t->nbuf; i++) {
    double x = t->buf[i];
    int j = i - 1;
    while (j >= 0 && t->buf[j] > x) { t->buf[j + 1] = t->buf[j]; j--; }
    t->buf[j + 1] = x;
  }
  /* merge the sorted centroid list with the sorted buffer, re-clustering
   * greedily under the k1 size bound */
  double nm[TD_NC + TD_BUF], nw[TD_NC + TD_BUF];
  int ni = 0, bi = 0, ci = 0;
  double total = t->total + (double)t->nbuf;
  double wSoFar = 0;
  double cm = 0, cw = 0;
  int have = 0;
  double qlimit = total * td_q(td_k(0) + 1.0);
  while (ci < t->nc || bi < t->nbuf) {
    double im, iw;
    if (ci < t->nc && (bi >= t->nbuf || t->mean[ci] <= t->buf[bi])) {
      im = t->mean[ci]; iw = t->w[ci]; ci++;
    } else {
      im = t->buf[bi]; iw = 1.0; bi++;
    }
    if (!have) { cm = im; cw = iw; have = 1; continue; }
    if (wSoFar + cw + iw <= qlimit) {
      cw += iw;
      cm += (im - cm) * iw / cw;     /* incremental weighted mean */
    } else {
      nm[ni] = cm; nw[ni] = cw; ni++;
      wSoFar += cw;
      qlimit = total * td_q(td_k(wSoFar / total) + 1.0);
      cm = im; cw = iw;
    }
  }
  if (have) { nm[ni] = cm; nw[ni] = cw; ni++; }
  if (ni > TD_NC) ni = TD_NC;        /* unreachable for COMP=100 */
  memcpy(t->mean, nm, (size_t)ni * sizeof(double));
  memcpy(t->w, nw, (size_t)ni * sizeof(double));
  t->nc = ni;
  t->total = total;
  t->nbuf = 0;
}

TD_FN void td_add(tdigest_t* t, double x) {
  if (isnan(x)) return;              /* reduceMappedRow skips NaN samples */
  if (x < t->tmin) t->tmin = x;
  if (x > t->tmax) t->tmax = x;
  t->buf[t->nbuf++] = x;
  if (t->nbuf == TD_BUF) td_flush(t);
}

TD_FN double td_quantile(tdigest_t* t, double q) {
  td_flush(t);
  if (t->nc == 0) return NAN;
  if (t->nc == 1) return t->mean[0];
  double index = q * t->total;
  if (index < t->w[0] / 2)           /* left tail: interpolate from min */
    return t->tmin + 2.0 * index / t->w[0] * (t->mean[0] - t->tmin);
  double wSoFar = 0;
  for (int i = 0; i + 1 < t->nc; i++) {
    double lo = wSoFar + t->w[i] / 2;
    double dw = (t->w[i] + t->w[i + 1]) / 2;
    if (index < lo + dw) {
      double z = (index - lo) / dw;
      return t->mean[i] * (1 - z) + t->mean[i + 1] * z;
    }
    wSoFar += t->w[i];
  }
  /* right tail: interpolate to max */
  double wl = t->w[t->nc - 1];
  double lo = t->total - wl / 2;
  double z = (index - lo) / (wl / 2);
  if (z < 0) z = 0;
  if (z > 1) z = 1;
  return t->mean[t->nc - 1] + z * (t->tmax - t->mean[t->nc - 1]);
}

#endif /* FDB_TDIGEST_IMPL_H */
