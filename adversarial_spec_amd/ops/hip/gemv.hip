// Decode GEMV: y[1,N] = x[1,K] @ W[K,N], bf16 in/out, fp32 accumulation.
//
// Batch-1 decode is bound by streaming the weight matrix once from HBM3E;
// hipBLASLt's batch-1 kernels measured 0.9-1.7 TB/s on this path (rocprof,
// profiles/), so the hot decode projections use this hand-written streamer:
// 16 B/lane coalesced weight reads with an 8-deep k unroll (8 independent
// loads in flight per thread), split-K fp32 partials for full-chip
// occupancy, and the cross-split combine FUSED into the same kernel via a
// last-block-arrives reduction (decode is kernel-count bound at ~450
// launches/token: a separate combine kernel costs ~5.5 us execution floor
// per GEMV, ~0.7 ms/token across the 33 projections of an 8B step).
//
// Tile: one block = 256 threads covers 64 output columns x a K-chunk.
//   thread t: vec-column (t % 8) (8 bf16 cols), k-lane (t / 8) of 32.
// Partials: [ksplit, N] fp32 in a per-stream scratch; the last block to
// finish a column-block sums them and writes bf16. The per-column-block
// atomic counters self-reset to zero so the buffers are reusable by the
// next launch on the same stream (and by every HIP-graph replay) with no
// zeroing pass.

#include "common.h"

extern "C" __global__ void __launch_bounds__(256)
gemv_fused_kernel(const ushort_t *__restrict__ x,
                  const ushort_t *__restrict__ w, float *__restrict__ part,
                  unsigned int *__restrict__ counters,
                  ushort_t *__restrict__ y, int K, int N, int kchunk,
                  int ksplit) {
  const int cb = blockIdx.x;   // column block (64 cols)
  const int ks = blockIdx.y;   // k split
  const int t = threadIdx.x;
  const int vc = t & 7;        // vec-column 0..7 (8 bf16 each)
  const int kl = t >> 3;       // k lane 0..31
  const int c0 = cb * 64;

  const int k0 = ks * kchunk;
  const int k1 = min(K, k0 + kchunk);

  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;

  // 8-deep k unroll: 8 independent 16 B weight loads in flight per thread
  // (4-deep measured latency-bound at ~47% of HBM roofline).
  int k = k0 + kl;
  for (; k + 224 < k1; k += 256) {
    bf16x8 wv[8];
    float xv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int kk = k + 32 * u;
      xv[u] = bf16_to_f32(x[kk]);
      wv[u] = ((const bf16x8 *)(w + (size_t)kk * N + c0))[vc];
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) {
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += xv[u] * bf16_to_f32(wv[u].u[j]);
    }
  }
  for (; k < k1; k += 32) {
    const float xv = bf16_to_f32(x[k]);
    const bf16x8 wv = ((const bf16x8 *)(w + (size_t)k * N + c0))[vc];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += xv * bf16_to_f32(wv.u[j]);
  }

  // reduce the 32 k-lanes per vec-column via LDS.
  __shared__ float red[256 * 8];
#pragma unroll
  for (int j = 0; j < 8; ++j) red[t * 8 + j] = acc[j];
  __syncthreads();

  if (ksplit == 1) {
    // single split: write the result directly, no workspace round-trip.
    if (t < 64) {
      const int vcc = t >> 3, j = t & 7;
      float sum = 0.f;
#pragma unroll 8
      for (int klane = 0; klane < 32; ++klane)
        sum += red[(vcc + 8 * klane) * 8 + j];
      y[c0 + vcc * 8 + j] = f32_to_bf16(sum);
    }
    return;
  }

  if (t < 64) {
    const int vcc = t >> 3, j = t & 7;
    float sum = 0.f;
#pragma unroll 8
    for (int klane = 0; klane < 32; ++klane)
      sum += red[(vcc + 8 * klane) * 8 + j];
    part[(size_t)ks * N + c0 + vcc * 8 + j] = sum;
  }
  __threadfence();  // publish this block's partials (per-wave stores)
  __syncthreads();

  // last block for this column-block combines all splits.
  __shared__ unsigned int arrival;
  if (t == 0) {
    arrival = atomicAdd(&counters[cb], 1u);
  }
  __syncthreads();
  if (arrival == (unsigned)(ksplit - 1)) {
    if (t == 0) counters[cb] = 0;  // self-reset for the next launch/replay
    __threadfence_block();
    if (t < 64) {
      const int col = c0 + t;
      float sum = 0.f;
      for (int s = 0; s < ksplit; ++s) sum += part[(size_t)s * N + col];
      y[col] = f32_to_bf16(sum);
    }
  }
}

extern "C" void launch_gemv(const ushort_t *x, const ushort_t *w, float *part,
                            unsigned int *counters, ushort_t *y, int K, int N,
                            int ksplit, hipStream_t stream) {
  const int ncb = N / 64;
  const int kchunk = (K + ksplit - 1) / ksplit;
  dim3 grid(ncb, ksplit);
  gemv_fused_kernel<<<grid, 256, 0, stream>>>(x, w, part, counters, y, K, N,
                                              kchunk, ksplit);
}
