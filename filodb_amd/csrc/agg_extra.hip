// Cross-series quantile (t-digest) and count_values aggregators on the GPU.
//
// Both run as presenters over the per-series [S×W] grid the scan writes
// (the two-phase reduce of DESIGN.md §4): one THREAD per (group, window)
// cell walks its group's members in ascending series order — the exact
// insertion order of the oracle's fold, and the reference's fastReduce
// accumulator order (AggrOverRangeVectors.scala:320-377), so the t-digest
// (shared implementation, tdigest_impl.h) produces bit-identical centroids
// to the oracle.
//
//   quantile:     QuantileRowAggregator.scala:21-76 (t-digest per cell,
//                 present step = digest.quantile(q))
//   count_values: CountValuesRowAggregator.scala:26-100 (value→frequency map
//                 per cell; > limit distinct values is an error — the
//                 reference throws at 1000)

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstring>
#include <cmath>

#include "chunk_format.h"
#include "scan_common.h"
#include "tdigest_impl.h"
#include "../../include/filodb_amd.h"

void fdb_set_error(const char* fmt, ...);   // chunk_builder.cpp

__global__ __launch_bounds__(64)
void quantile_cell_kernel(const double* __restrict__ grid,
                          const int32_t* __restrict__ sbg,
                          const int32_t* __restrict__ goff,
                          int ng, int nw, double q,
                          double* __restrict__ out) {
  size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (size_t)ng * nw) return;
  int g = (int)(idx / nw), w = (int)(idx % nw);
  tdigest_t td;
  td_init(&td);
  for (int i = goff[g]; i < goff[g + 1]; i++)
    td_add(&td, grid[(size_t)sbg[i] * nw + w]);
  out[idx] = td_quantile(&td, q);
}

#define CV_CAP 1000      // the reference's hard distinct-value limit

__global__ __launch_bounds__(64)
void count_values_kernel(const double* __restrict__ grid,
                         const int32_t* __restrict__ sbg,
                         const int32_t* __restrict__ goff,
                         int ng, int nw, int k_cap,
                         double* __restrict__ out_vals,
                         double* __restrict__ out_cnts,
                         int32_t* __restrict__ out_n,
                         int32_t* __restrict__ overflow) {
  size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (size_t)ng * nw) return;
  int g = (int)(idx / nw), w = (int)(idx % nw);
  double cv[CV_CAP], cc[CV_CAP];
  int n = 0;
  for (int i = goff[g]; i < goff[g + 1]; i++) {
    double x = grid[(size_t)sbg[i] * nw + w];
    if (isnan(x)) continue;
    int lo = 0, hi = n;
    while (lo < hi) { int mid = (lo + hi) / 2;
      if (cv[mid] < x) lo = mid + 1; else hi = mid; }
    if (lo < n && cv[lo] == x) { cc[lo] += 1; }
    else {
      if (n >= k_cap) { atomicExch(overflow, 1); n = -1; break; }
      for (int j = n; j > lo; j--) { cv[j] = cv[j - 1]; cc[j] = cc[j - 1]; }
      cv[lo] = x; cc[lo] = 1;
      n++;
    }
  }
  out_n[idx] = n < 0 ? 0 : n;
  for (int j = 0; j < n; j++) {
    out_vals[idx * (size_t)k_cap + j] = cv[j];
    out_cnts[idx * (size_t)k_cap + j] = cc[j];
  }
}

int32_t fdb_launch_quantile_cells(hipStream_t stream, const double* grid,
                                  const int32_t* sbg, const int32_t* goff,
                                  int ng, int nw, double q, double* out) {
  size_t cells = (size_t)ng * nw;
  hipLaunchKernelGGL(quantile_cell_kernel,
                     dim3((uint32_t)((cells + 63) / 64)), dim3(64), 0, stream,
                     grid, sbg, goff, ng, nw, q, out);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    fdb_set_error("quantile_cell_kernel launch failed: %s", hipGetErrorString(e));
    return FDB_ERR;
  }
  return FDB_OK;
}

int32_t fdb_launch_count_values(hipStream_t stream, const double* grid,
                                const int32_t* sbg, const int32_t* goff,
                                int ng, int nw, int k_cap,
                                double* out_vals, double* out_cnts,
                                int32_t* out_n, int32_t* overflow) {
  if (k_cap < 1 || k_cap > CV_CAP) {
    fdb_set_error("count_values k_cap %d out of range 1..%d", k_cap, CV_CAP);
    return FDB_ERR_BADARG;
  }
  size_t cells = (size_t)ng * nw;
  hipLaunchKernelGGL(count_values_kernel,
                     dim3((uint32_t)((cells + 63) / 64)), dim3(64), 0, stream,
                     grid, sbg, goff, ng, nw, k_cap, out_vals, out_cnts,
                     out_n, overflow);
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) {
    fdb_set_error("count_values_kernel launch failed: %s", hipGetErrorString(e));
    return FDB_ERR;
  }
  return FDB_OK;
}
