// MFMA throughput microbench: bf16 16x16x32 vs fp8 (e4m3) 16x16x32.
//
// Decides BASELINE config 5's "fp8 MFMA prefill attention" question with
// a measurement (round-2 verdict item 5): gfx950's NON-scaled fp8 MFMA
// (mfma_f32_16x16x32_fp8_fp8) shares the bf16 issue rate — only the
// block-scaled MX K=128 forms reach the 2x fp8 peak, and those impose a
// K=128 fragment layout that the online-softmax QK^T/PV structure cannot
// use per 32-element K-step. If this probe shows ~1.0x, fp8 attention
// buys no compute rate on this chip and only halves K/V staging bytes
// (not the prefill bound), justifying the bf16 attention path.
//
// Each wave issues DEPTH independent-accumulator MFMAs in a loop; the
// kernel is pure matrix-pipe issue (no memory in the loop).

#include "common.h"

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8v;
typedef __attribute__((__vector_size__(2 * sizeof(int)))) int i32x2v;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4v;

#define RATE_ITERS 4096
#define RATE_ACCS 8  // independent accumulators cover the dependent latency

extern "C" __global__ void __launch_bounds__(256)
mfma_rate_bf16_kernel(const ushort_t *__restrict__ seed,
                      float *__restrict__ out) {
  bf16x8v a = *(const bf16x8v *)(seed + (threadIdx.x & 63) * 8);
  bf16x8v b = *(const bf16x8v *)(seed + 512 + (threadIdx.x & 63) * 8);
  f32x4v acc[RATE_ACCS];
#pragma unroll
  for (int i = 0; i < RATE_ACCS; ++i) acc[i] = (f32x4v){0, 0, 0, 0};
  for (int it = 0; it < RATE_ITERS; ++it) {
#pragma unroll
    for (int i = 0; i < RATE_ACCS; ++i)
      acc[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[i], 0, 0, 0);
  }
  float s = 0;
#pragma unroll
  for (int i = 0; i < RATE_ACCS; ++i) s += acc[i][0] + acc[i][3];
  if (s == 12345.0f) out[blockIdx.x] = s;  // keep the loop live
}

extern "C" __global__ void __launch_bounds__(256)
mfma_rate_fp8_kernel(const ushort_t *__restrict__ seed,
                     float *__restrict__ out) {
  // fp8 operands are 8 bytes per lane (8 x e4m3 for K=32)
  i32x2v a = *(const i32x2v *)(seed + (threadIdx.x & 63) * 8);
  i32x2v b = *(const i32x2v *)(seed + 512 + (threadIdx.x & 63) * 8);
  long la = ((long)a[1] << 32) | (unsigned)a[0];
  long lb = ((long)b[1] << 32) | (unsigned)b[0];
  f32x4v acc[RATE_ACCS];
#pragma unroll
  for (int i = 0; i < RATE_ACCS; ++i) acc[i] = (f32x4v){0, 0, 0, 0};
  for (int it = 0; it < RATE_ITERS; ++it) {
#pragma unroll
    for (int i = 0; i < RATE_ACCS; ++i)
      acc[i] =
          __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(la, lb, acc[i], 0, 0, 0);
  }
  float s = 0;
#pragma unroll
  for (int i = 0; i < RATE_ACCS; ++i) s += acc[i][0] + acc[i][3];
  if (s == 12345.0f) out[blockIdx.x] = s;
}

extern "C" void launch_mfma_rate(const ushort_t *seed, float *out, int which,
                                 int blocks, hipStream_t stream) {
  if (which == 8) {
    mfma_rate_fp8_kernel<<<blocks, 256, 0, stream>>>(seed, out);
  } else {
    mfma_rate_bf16_kernel<<<blocks, 256, 0, stream>>>(seed, out);
  }
}
