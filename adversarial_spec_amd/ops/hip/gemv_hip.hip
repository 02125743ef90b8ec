#include "hip/hip_runtime.h"
// Decode GEMV: y[1,N] = x[1,K] @ W[K,N], bf16 in/out, fp32 accumulation.
//
// Batch-1 decode is bound by streaming the weight matrix once from HBM3E;
// hipBLASLt's batch-1 kernels measured 0.9-1.7 TB/s on this path (rocprof,
// profiles/), so the hot decode projections use this hand-written streamer
// instead: 16 B/lane coalesced weight reads, split-K fp32 partials for
// full-chip occupancy (>= ~1024 workgroups), and a tiny combine kernel.
//
// Tile: one block = 256 threads covers 64 output columns x a K-chunk.
//   thread t: vec-column (t % 8) (8 bf16 cols), k-lane (t / 8) of 32.
// Partials: [ksplit, N] fp32; combine sums and converts to bf16.

#include "common.h"

extern "C" __global__ void __launch_bounds__(256)
gemv_partial_kernel(const ushort_t *__restrict__ x,
                    const ushort_t *__restrict__ w, float *__restrict__ part,
                    int K, int N, int kchunk, long x_stride_unused) {
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

  // 4-deep k unroll: 4 independent 16 B weight loads in flight per thread
  // (single-load version measured latency-bound at ~2 TB/s).
  int k = k0 + kl;
  for (; k + 96 < k1; k += 128) {
    bf16x8 wv[4];
    float xv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int kk = k + 32 * u;
      xv[u] = bf16_to_f32(x[kk]);
      wv[u] = ((const bf16x8 *)(w + (size_t)kk * N + c0))[vc];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
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

  // reduce the 32 k-lanes per vec-column.
  // lanes within a wave: t = vc + 8*kl -> same vc every 8 threads; a wave
  // holds kl 0..7 for its vcs. Cross-wave via LDS.
  __shared__ float red[256 * 8];
#pragma unroll
  for (int j = 0; j < 8; ++j) red[t * 8 + j] = acc[j];
  __syncthreads();
  if (t < 64) {
    // thread t covers one (vec-column, element) pair: vc = t/8, j = t%8
    const int vcc = t >> 3, j = t & 7;
    float sum = 0.f;
#pragma unroll 8
    for (int klane = 0; klane < 32; ++klane) {
      sum += red[(vcc + 8 * klane) * 8 + j];
    }
    part[(size_t)ks * N + c0 + vcc * 8 + j] = sum;
  }
}

extern "C" __global__ void __launch_bounds__(256)
gemv_combine_kernel(const float *__restrict__ part, ushort_t *__restrict__ y,
                    int N, int ksplit) {
  for (int n = blockIdx.x * blockDim.x + threadIdx.x; n < N;
       n += gridDim.x * blockDim.x) {
    float s = 0.f;
    for (int ks = 0; ks < ksplit; ++ks) s += part[(size_t)ks * N + n];
    y[n] = f32_to_bf16(s);
  }
}

extern "C" void launch_gemv(const ushort_t *x, const ushort_t *w, float *part,
                            ushort_t *y, int K, int N, int ksplit,
                            hipStream_t stream) {
  const int ncb = N / 64;
  const int kchunk = (K + ksplit - 1) / ksplit;
  dim3 grid(ncb, ksplit);
 hipLaunchKernelGGL(( gemv_partial_kernel), dim3(grid), dim3(256), 0, stream, x, w, part, K, N, kchunk, 0);
 hipLaunchKernelGGL(( gemv_combine_kernel), dim3(dim3(min(256, (N + 255) / 256))), dim3(256), 0, stream, 
      part, y, N, ksplit);
}
