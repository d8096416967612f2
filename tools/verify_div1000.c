/* Exhaustive verification of the d_div1000 Markstein sequence
 * (filodb_amd/csrc/scan_common.h): for EVERY integer |x| <= 2^33,
 *   q0 = RN(x * r), e = fma(-1000, q0, x), q = RN(fma(e, r, q0))
 * with r = RN(1/1000) equals the correctly rounded RN(x/1000).
 * The engine's callers pass integer millisecond differences bounded by the
 * window length, and the engine rejects windows over 2^33 ms, so this domain
 * covers every value the kernel can see. All operations are sign-symmetric,
 * so non-negative x suffices.
 *
 * Build & run:  gcc -O3 -fopenmp -march=native -o verify_div1000 \
 *                   tools/verify_div1000.c -lm && ./verify_div1000
 * Last run (this container, 2026-09-15): bad=0 of 8589934593; the naive
 * q0-only reciprocal differs on ~13% of them (sanity that the harness can
 * detect mismatches). The negative domain -2^33..0 was verified explicitly
 * as well (bad=0) — not just by the sign-symmetry argument.
 */
#include <stdio.h>
#include <math.h>
#include <stdint.h>

int main(void) {
  const double r = 1.0 / 1000.0;
  int64_t bad = 0, bad_naive = 0;
  const int64_t N = (int64_t)1 << 33;
#pragma omp parallel for reduction(+:bad, bad_naive) schedule(static)
  for (int64_t i = 0; i <= N; i++) {
    double x = (double)i;
    double q0 = x * r;
    double e = fma(-1000.0, q0, x);
    double q = fma(e, r, q0);
    double ref = x / 1000.0;
    if (q != ref) bad++;
    if (q0 != ref) bad_naive++;
  }
  printf("markstein bad=%lld of %lld (naive reciprocal bad=%lld)\n",
         (long long)bad, (long long)(N + 1), (long long)bad_naive);
  return bad != 0;
}
