// Decode GEMV: y[1,N] = x[1,K] @ W[K,N], bf16 in/out, fp32 accumulation.
//
// Batch-1 decode is bound by streaming the weight matrix once from HBM3E;
// hipBLASLt's batch-1 kernels measured 0.9-1.7 TB/s on this path (rocprof,
// profiles/), so the hot decode projections use this hand-written streamer:
// 16 B/lane coalesced weight reads with an 8-deep k unroll (8 independent
// loads in flight per thread) and a DIRECT bf16 write — no split-K.
//
// Why no split-K: a cross-block combine (either a second kernel or a
// last-block-arrives reduction) was measured strictly worse on this chip:
// the separate combine kernel costs a ~5.5 us execution floor per GEMV
// (~0.7 ms/token over 33 projections), and the fused last-block variant
// needs device-scope fences, which on the 8-XCD MI355X trigger cross-L2
// traffic that slowed the WHOLE device ~5x (profiles/, round 1). Instead
// each block owns 64 output columns over the FULL K; the debate engine
// runs 3+ co-resident opponents on separate HIP streams, so the chip is
// filled by opponent-level concurrency rather than intra-GEMV splits.
//
// Tile: one block = 256 threads covers 64 output columns x all of K.
//   thread t: vec-column (t % 8) (8 bf16 cols), k-lane (t / 8) of 32.

#include "common.h"

extern "C" __global__ void __launch_bounds__(256)
gemv_kernel(const ushort_t *__restrict__ x, const ushort_t *__restrict__ w,
            ushort_t *__restrict__ y, int K, int N) {
  const int cb = blockIdx.x;   // column block (64 cols)
  const int t = threadIdx.x;
  const int vc = t & 7;        // vec-column 0..7 (8 bf16 each)
  const int kl = t >> 3;       // k lane 0..31
  const int c0 = cb * 64;

  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;

  // 8-deep k unroll: 8 independent 16 B weight loads in flight per thread
  // (4-deep measured latency-bound at ~47% of HBM roofline).
  int k = kl;
  for (; k + 224 < K; k += 256) {
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
  for (; k < K; k += 32) {
    const float xv = bf16_to_f32(x[k]);
    const bf16x8 wv = ((const bf16x8 *)(w + (size_t)k * N + c0))[vc];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += xv * bf16_to_f32(wv.u[j]);
  }

  // reduce the 32 k-lanes per vec-column via LDS, then write bf16 directly.
  __shared__ float red[256 * 8];
#pragma unroll
  for (int j = 0; j < 8; ++j) red[t * 8 + j] = acc[j];
  __syncthreads();
  if (t < 64) {
    const int vcc = t >> 3, j = t & 7;
    float sum = 0.f;
#pragma unroll 8
    for (int klane = 0; klane < 32; ++klane)
      sum += red[(vcc + 8 * klane) * 8 + j];
    y[c0 + vcc * 8 + j] = f32_to_bf16(sum);
  }
}

extern "C" void launch_gemv(const ushort_t *x, const ushort_t *w, ushort_t *y,
                            int K, int N, hipStream_t stream) {
  gemv_kernel<<<dim3(N / 64), 256, 0, stream>>>(x, w, y, K, N);
}
