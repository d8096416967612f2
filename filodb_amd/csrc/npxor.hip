// Gorilla-XOR double bitstream decode on the GPU (NibblePack.unpackDoubleXOR,
// NibblePack.scala:360-394) — the wavefront ballot/prefix design the
// north_star names: one wave per stream; each iteration the lanes
//
//  1. walk 8 group headers on uniform staged bytes (bitmask byte → ballot of
//     present values; nibble-width byte → group length), producing the 8
//     group offsets,
//  2. extract 64 packed XOR deltas in parallel (lane = one value: 8 groups ×
//     8 slots), and
//  3. reconstruct the chain with a wave-wide inclusive PREFIX-XOR (XOR is
//     associative — value[i] = first ^ delta[1] ^ … ^ delta[i]), carrying
//     lane 63's word into the next iteration.
//
// The streams decode bit-exactly against the host/oracle decoder (and
// roundtrip against packDoubles); 64 doubles retire per prefix pass.

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstring>

#include "chunk_format.h"
#include "scan_common.h"
#include "../../include/filodb_amd.h"

void fdb_set_error(const char* fmt, ...);   // chunk_builder.cpp

__device__ __forceinline__ uint64_t wave_incl_xor(uint64_t x, int lane) {
  for (int off = 1; off < 64; off <<= 1) {
    uint64_t t = __shfl_up(x, off);
    if (lane >= off) x ^= t;
  }
  return x;
}

// streams: concatenated packed bytes; offs[i]/lens[i] describe stream i,
// counts[i] its decoded length; outs land at out + out_offs[i]
__global__ __launch_bounds__(256)
void xor_unpack_kernel(const uint8_t* __restrict__ blob,
                       const int64_t* __restrict__ offs,
                       const int32_t* __restrict__ counts,
                       const int64_t* __restrict__ out_offs,
                       int num_streams,
                       double* __restrict__ out) {
  const int wave = __builtin_amdgcn_readfirstlane(threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;

  for (int s = blockIdx.x * 4 + wave; s < num_streams; s += gridDim.x * 4) {
    const uint8_t* p = blob + offs[s];
    const int n = counts[s];
    double* o = out + out_offs[s];
    if (n < 1) continue;
    uint64_t carry = d_i64(p);               // first double stored raw
    if (lane == 0) memcpy(&o[0], &carry, 8);
    int pos = 8, i = 1;
    while (i < n) {
      // header walk for up to 8 groups (wave-uniform; staged 256 B)
      const uint8_t* gp = p + pos;
      uint32_t stg = estream_stage(gp, lane);
      const int shift = (int)((uintptr_t)gp & 3);
      int goff[8] = {0}, gbits[8] = {0}, gtrail[8] = {0};
      uint32_t gmask[8] = {0};
      int off = 0;
      int ngroups = 0;
      for (int g = 0; g < 8 && i + ngroups * 8 < n + 7; g++) {
        uint32_t mask = estream_byte_uni(true, stg, shift, gp, off);
        int numBits = 0, trail = 0, glen;
        if (mask == 0) {
          glen = 1;
        } else {
          int widths = (int)estream_byte_uni(true, stg, shift, gp, off + 1);
          numBits = ((widths >> 4) + 1) * 4;
          trail = (widths & 0x0f) * 4;
          glen = 2 + (numBits * __popc(mask) + 7) / 8;
        }
        if (off + glen > 244) break;   // k+8+shift stays under the 64-dword stage
        goff[g] = off; gbits[g] = numBits; gtrail[g] = trail; gmask[g] = mask;
        off += glen;
        ngroups++;
        if (i + ngroups * 8 >= n) break;      // enough values decoded
      }
      if (ngroups == 0) {                     // giant group: bounce via lane 0
        uint32_t mask = estream_byte(true, stg, shift, gp, 0);
        int widths = (int)estream_byte(true, stg, shift, gp, 1);
        int numBits = ((widths >> 4) + 1) * 4;
        int trail = (widths & 0x0f) * 4;
        int glen = 2 + (numBits * __popc(mask) + 7) / 8;
        int k = lane & 7;
        uint64_t delta = 0;
        if ((lane >> 3) == 0 && (mask & (1u << k))) {
          int slot = __popc(mask & ((1u << k) - 1));
          int bitpos = slot * numBits;
          uint64_t w;
          memcpy(&w, gp + 2 + (bitpos >> 3), 8);
          w >>= (bitpos & 7);
          if (numBits > 64 - (bitpos & 7)) {
            uint64_t hi = gp[2 + (bitpos >> 3) + 8];
            w |= hi << (64 - (bitpos & 7));
          }
          uint64_t m = numBits >= 64 ? ~0ULL : ((1ULL << numBits) - 1);
          delta = (w & m) << trail;
        }
        uint64_t chain = wave_incl_xor(lane < 8 ? delta : 0, lane) ^ carry;
        if (lane < 8 && i + lane < n) memcpy(&o[i + lane], &chain, 8);
        int take = n - i < 8 ? n - i : 8;
        carry = __shfl(chain, take - 1 < 7 ? take - 1 : 7);
        i += take;
        pos += glen;
        continue;
      }
      // lane = (group, slot): extract this lane's delta. The estream reads
      // are SHUFFLES, so every lane must execute them (inactive source lanes
      // yield undefined data) — only the final delta is predicated.
      const int g = lane >> 3, k = lane & 7;
      const bool lvalid = g < ngroups && (gmask[g] & (1u << k));
      int slot = lvalid ? __popc(gmask[g] & ((1u << k) - 1)) : 0;
      int bitpos = slot * gbits[g];
      int koff = goff[g] + 2 + (bitpos >> 3);
      uint64_t w = estream_w64(true, stg, shift, gp, koff);
      uint32_t b8 = estream_byte(true, stg, shift, gp, koff + 8);
      uint64_t delta = 0;
      if (lvalid) {
        int sh = bitpos & 7;
        uint64_t v = w >> sh;
        if (gbits[g] > 64 - sh) v |= (uint64_t)b8 << (64 - sh);
        uint64_t m = gbits[g] >= 64 ? ~0ULL : ((1ULL << gbits[g]) - 1);
        delta = (v & m) << gtrail[g];
      }
      // prefix-XOR reconstructs up to 64 chain values at once
      uint64_t chain = wave_incl_xor(g < ngroups ? delta : 0, lane) ^ carry;
      const int avail = ngroups * 8;
      const int take = (n - i) < avail ? (n - i) : avail;
      if (lane < take) memcpy(&o[i + lane], &chain, 8);
      carry = __shfl(chain, take - 1);
      i += take;
      pos += off;
    }
    d_wait_lds();
    __builtin_amdgcn_wave_barrier();
  }
}

hipStream_t fdb_engine_stream(fdb_engine_t* e);   // engine.hip

// host entry: decodes num_streams packed streams (host memory) on the GPU.
// offs/counts/out_offs are host arrays; out is a host buffer.
extern "C" int32_t fdb_gpu_unpack_doubles_xor(fdb_engine_t* e,
                                              const uint8_t* blob, int64_t blob_len,
                                              const int64_t* offs,
                                              const int32_t* counts,
                                              const int64_t* out_offs,
                                              int32_t num_streams,
                                              double* out, int64_t out_len) {
  hipStream_t stream = fdb_engine_stream(e);
  uint8_t* db = nullptr;
  int64_t *doffs = nullptr, *dooffs = nullptr;
  int32_t* dcounts = nullptr;
  double* dout = nullptr;
  int32_t rc = FDB_ERR;
  // +256 B tail pad so the wave staging never reads past the blob
  if (hipMalloc(&db, (size_t)blob_len + 256) != hipSuccess) goto done;
  if (hipMemset(db + blob_len, 0, 256) != hipSuccess) goto done;
  if (hipMemcpy(db, blob, (size_t)blob_len, hipMemcpyHostToDevice) != hipSuccess) goto done;
  if (hipMalloc(&doffs, (size_t)num_streams * 8) != hipSuccess) goto done;
  if (hipMalloc(&dcounts, (size_t)num_streams * 4) != hipSuccess) goto done;
  if (hipMalloc(&dooffs, (size_t)num_streams * 8) != hipSuccess) goto done;
  if (hipMalloc(&dout, (size_t)out_len * 8) != hipSuccess) goto done;
  if (hipMemcpy(doffs, offs, (size_t)num_streams * 8, hipMemcpyHostToDevice) != hipSuccess) goto done;
  if (hipMemcpy(dcounts, counts, (size_t)num_streams * 4, hipMemcpyHostToDevice) != hipSuccess) goto done;
  if (hipMemcpy(dooffs, out_offs, (size_t)num_streams * 8, hipMemcpyHostToDevice) != hipSuccess) goto done;
  {
    int grid = (num_streams + 3) / 4;
    if (grid > 8192) grid = 8192;
    if (grid < 1) grid = 1;
    hipLaunchKernelGGL(xor_unpack_kernel, dim3(grid), dim3(256), 0, stream,
                       db, doffs, dcounts, dooffs, num_streams, dout);
    if (hipGetLastError() != hipSuccess) {
      fdb_set_error("xor_unpack_kernel launch failed");
      goto done;
    }
  }
  if (hipStreamSynchronize(stream) != hipSuccess) goto done;
  if (hipMemcpy(out, dout, (size_t)out_len * 8, hipMemcpyDeviceToHost) != hipSuccess) goto done;
  rc = FDB_OK;
done:
  if (rc != FDB_OK && rc == FDB_ERR) fdb_set_error("xor decode device error");
  (void)hipFree(db); (void)hipFree(doffs); (void)hipFree(dcounts);
  (void)hipFree(dooffs); (void)hipFree(dout);
  return rc;
}
