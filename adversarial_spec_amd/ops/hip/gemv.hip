// Decode GEMV: y[1,N] = x[1,K] @ W[N,K]^T, bf16 in/out, fp32 accumulation.
//
// Weights are ROW-MAJOR [out, in] (HF layout; same convention as the
// MFMA GEMM), so each output element is a contiguous row dot-product —
// the weight matrix streams once from HBM3E in full 256 B row segments.
//
// Tiling: block = 256 threads (4 waves); each wave owns 4 output rows via
// 16-lane groups; a lane reads 16 B (8 bf16) per k-step, 4-deep unrolled
// so >= 4 independent loads are in flight per lane. Cross-lane reduce via
// 16-lane shfl; no split-K, no workspace (the debate engine runs 3+
// co-resident opponents on separate HIP streams, so the chip is filled by
// opponent-level concurrency — and a cross-block combine needs
// device-scope fences that thrash cross-XCD L2, measured ~5x whole-device
// slowdown in round 1).

#include "common.h"
// dot8_bf16 (packed v_dot2_f32_bf16 row dot) comes from common.h

// 16-lane-group sum (lanes p, p+1, .., p+15 with stride 1)
DEVINL float group16_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

extern "C" __global__ void __launch_bounds__(256)
gemv_kernel(const ushort_t *__restrict__ x, const ushort_t *__restrict__ w,
            ushort_t *__restrict__ y, int K, int N) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int rg = lane >> 4;       // row group within wave: 0..3
  const int sl = lane & 15;       // k-slice lane: 0..15
  const int n = blockIdx.x * 16 + wid * 4 + rg;
  if (n >= N) return;

  const ushort_t *wr = w + (size_t)n * K;
  const int nc = K / 8;           // 16 B chunks per row (K % 8 == 0)

  float acc = 0.f;
  int c = sl;
  // 4-deep unroll (this kernel only runs at large N where the grid itself
  // saturates HBM; 8-deep measured slower there)
  for (; c + 48 < nc; c += 64) {
    bf16x8 wv[4], xv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      wv[u] = ((const bf16x8 *)wr)[c + 16 * u];
      xv[u] = ((const bf16x8 *)x)[c + 16 * u];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) acc = dot8_bf16(xv[u], wv[u], acc);
  }
  for (; c < nc; c += 16) {
    const bf16x8 wv = ((const bf16x8 *)wr)[c];
    const bf16x8 xv = ((const bf16x8 *)x)[c];
    acc = dot8_bf16(xv, wv, acc);
  }

  const float sum = group16_sum(acc);
  if (sl == 0) y[n] = f32_to_bf16(sum);
}

// 32-lanes-per-row variant for SMALL N (o/qkv-sized): the 16-lane kernel
// leaves only ~1 wave/SIMD there, so halve the rows per wave and double
// both the k-parallelism per row and the block count.
extern "C" __global__ void __launch_bounds__(256)
gemv_kernel_w32(const ushort_t *__restrict__ x, const ushort_t *__restrict__ w,
                ushort_t *__restrict__ y, int K, int N) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int rg = lane >> 5;       // row group within wave: 0..1
  const int sl = lane & 31;       // k-slice lane: 0..31
  const int n = blockIdx.x * 8 + wid * 2 + rg;
  if (n >= N) return;

  const ushort_t *wr = w + (size_t)n * K;
  const int nc = K / 8;

  float acc = 0.f;
  int c = sl;
  for (; c + 224 < nc; c += 256) {
    bf16x8 wv[8], xv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      wv[u] = ((const bf16x8 *)wr)[c + 32 * u];
      xv[u] = ((const bf16x8 *)x)[c + 32 * u];
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) acc = dot8_bf16(xv[u], wv[u], acc);
  }
  for (; c < nc; c += 32) {
    const bf16x8 wv = ((const bf16x8 *)wr)[c];
    const bf16x8 xv = ((const bf16x8 *)x)[c];
    acc = dot8_bf16(xv, wv, acc);
  }

#pragma unroll
  for (int off = 16; off > 0; off >>= 1) acc += __shfl_xor(acc, off, WAVE);
  if (sl == 0) y[n] = f32_to_bf16(acc);
}

extern "C" void launch_gemv(const ushort_t *x, const ushort_t *w, ushort_t *y,
                            int K, int N, hipStream_t stream) {
  if (N <= 8192) {
    gemv_kernel_w32<<<dim3((N + 7) / 8), 256, 0, stream>>>(x, w, y, K, N);
  } else {
    gemv_kernel<<<dim3((N + 15) / 16), 256, 0, stream>>>(x, w, y, K, N);
  }
}

// Fused gate_up GEMV + SwiGLU for the decode path:
//   act[n] = silu(x @ Wg[n]) * (x @ Wu[n]),  W = [gate rows | up rows].
// Each 32-lane group computes BOTH row dots for one n (same x chunks),
// so the separate swiglu launch (+ the gu activation round-trip)
// disappears from the ~330-kernel decode step. Weight bytes unchanged —
// this is a launch/boundary fusion, not a traffic change.
extern "C" __global__ void __launch_bounds__(256)
gemv_gateup_kernel(const ushort_t *__restrict__ x,
                   const ushort_t *__restrict__ w,
                   ushort_t *__restrict__ act, int K, int F) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int rg = lane >> 5;       // row group within wave: 0..1
  const int sl = lane & 31;       // k-slice lane: 0..31
  const int n = blockIdx.x * 8 + wid * 2 + rg;
  if (n >= F) return;

  const ushort_t *wg = w + (size_t)n * K;
  const ushort_t *wu = w + (size_t)(n + F) * K;
  const int nc = K / 8;

  float ag = 0.f, au = 0.f;
  int c = sl;
  for (; c + 96 < nc; c += 128) {
    bf16x8 gv[4], uv[4], xv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      gv[u] = ((const bf16x8 *)wg)[c + 32 * u];
      uv[u] = ((const bf16x8 *)wu)[c + 32 * u];
      xv[u] = ((const bf16x8 *)x)[c + 32 * u];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      ag = dot8_bf16(xv[u], gv[u], ag);
      au = dot8_bf16(xv[u], uv[u], au);
    }
  }
  for (; c < nc; c += 32) {
    const bf16x8 xv = ((const bf16x8 *)x)[c];
    ag = dot8_bf16(xv, ((const bf16x8 *)wg)[c], ag);
    au = dot8_bf16(xv, ((const bf16x8 *)wu)[c], au);
  }

#pragma unroll
  for (int off = 16; off > 0; off >>= 1) {
    ag += __shfl_xor(ag, off, WAVE);
    au += __shfl_xor(au, off, WAVE);
  }
  if (sl == 0) {
    const float s = ag / (1.0f + __expf(-ag));  // silu(gate)
    act[n] = f32_to_bf16(s * au);
  }
}

extern "C" void launch_gemv_gateup(const ushort_t *x, const ushort_t *w,
                                   ushort_t *act, int K, int F,
                                   hipStream_t stream) {
  gemv_gateup_kernel<<<dim3((F + 7) / 8), 256, 0, stream>>>(x, w, act, K, F);
}

// ---- norm/residual-fused decode GEMVs -----------------------------------
//
// Decode-step anatomy (profiles/r02_measurements.md): add_rmsnorm x2 is
// 0.30 ms/tok of purely launch-bound small kernels (64 x ~4.7 us), plus a
// share of the graph's kernel-boundary gap. Both halves fuse into the
// GEMVs that surround them:
//   - the rmsnorm SCALE commutes out of the dot product:
//       y[n] = rms(x) * sum_i x[i]*wln[i]*W[n,i]
//     so the CONSUMING GEMV accumulates the wln-weighted dot and sum(x^2)
//     from the x chunks its lane group already streams (VALU is nearly
//     idle at the HBM weight-streaming roofline), applying one rsqrt per
//     output element after the cross-lane reduce;
//   - the residual ADD is a one-element epilogue on the PRODUCING GEMV
//     (resid[n] += dot), in place: each n is read and written by exactly
//     one lane group, and the stream serializes it against the consumer.
// Eliminates 65 launches/step for the 8B decode (2/layer + first norm);
// the TP decode path keeps the unfused sequence (the all-reduce must see
// the RAW partial projection before the residual add).

// Cooperative prologue shared by the norm-fused kernels: the whole block
// computes xl = x*wln into LDS (bf16) and block-reduces sum(x^2), so the
// weight-streaming main loop is EXACTLY the plain GEMV's 8-deep dot8_bf16
// loop with its x reads retargeted at LDS. A first cut that interleaved
// x/wln VMEM loads + f32 unpack math into the weight loop measured the
// step ~0.4 ms SLOWER than unfused (the mixed-stream loop loses the
// weight-load latency hiding; ds_read does not occupy a vmcnt slot, LDS
// staging restores it). Returns rms; xl[] is barrier-visible on return.
DEVINL float norm_stage_lds(const ushort_t *__restrict__ x,
                            const ushort_t *__restrict__ wln,
                            ushort_t *xl, float *red, int K, float eps) {
  float s2 = 0.f;
  const int nc = K / 8;
  for (int i = threadIdx.x; i < nc; i += 256) {
    const f32x8 xf = unpack8(((const bf16x8 *)x)[i]);
    const f32x8 lf = unpack8(((const bf16x8 *)wln)[i]);
    f32x8 p;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      p.v[j] = xf.v[j] * lf.v[j];
      s2 = fmaf(xf.v[j], xf.v[j], s2);
    }
    ((bf16x8 *)xl)[i] = pack8(p);
  }
  // block_reduce_sum's first barrier also makes the xl writes visible
  s2 = block_reduce_sum(s2, red);
  return rsqrtf(s2 / K + eps);
}

extern "C" __global__ void __launch_bounds__(256)
gemv_norm_w32_kernel(const ushort_t *__restrict__ x,
                     const ushort_t *__restrict__ wln,
                     const ushort_t *__restrict__ w,
                     ushort_t *__restrict__ y, int K, int N, float eps) {
  extern __shared__ ushort_t xl[];  // K bf16: x*wln (unnormalized)
  __shared__ float red[16];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int rg = lane >> 5;
  const int sl = lane & 31;
  const int n0 = blockIdx.x * 8 + wid * 2 + rg;
  // no early return before the barrier: clamp the row, guard the write
  const int n = n0 < N ? n0 : N - 1;

  const float rms = norm_stage_lds(x, wln, xl, red, K, eps);

  const ushort_t *wr = w + (size_t)n * K;
  const int nc = K / 8;
  float acc = 0.f;
  int c = sl;
  for (; c + 224 < nc; c += 256) {
    bf16x8 wv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) wv[u] = ((const bf16x8 *)wr)[c + 32 * u];
#pragma unroll
    for (int u = 0; u < 8; ++u)
      acc = dot8_bf16(((const bf16x8 *)xl)[c + 32 * u], wv[u], acc);
  }
  for (; c < nc; c += 32)
    acc = dot8_bf16(((const bf16x8 *)xl)[c], ((const bf16x8 *)wr)[c], acc);

#pragma unroll
  for (int off = 16; off > 0; off >>= 1) acc += __shfl_xor(acc, off, WAVE);
  if (sl == 0 && n0 < N) y[n0] = f32_to_bf16(acc * rms);
}

// 16-lane-group norm variant for LARGE N (lm_head: final_norm fused)
extern "C" __global__ void __launch_bounds__(256)
gemv_norm_kernel(const ushort_t *__restrict__ x,
                 const ushort_t *__restrict__ wln,
                 const ushort_t *__restrict__ w,
                 ushort_t *__restrict__ y, int K, int N, float eps) {
  extern __shared__ ushort_t xl[];
  __shared__ float red[16];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int rg = lane >> 4;
  const int sl = lane & 15;
  const int n0 = blockIdx.x * 16 + wid * 4 + rg;
  const int n = n0 < N ? n0 : N - 1;

  const float rms = norm_stage_lds(x, wln, xl, red, K, eps);

  const ushort_t *wr = w + (size_t)n * K;
  const int nc = K / 8;
  float acc = 0.f;
  int c = sl;
  for (; c + 48 < nc; c += 64) {
    bf16x8 wv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) wv[u] = ((const bf16x8 *)wr)[c + 16 * u];
#pragma unroll
    for (int u = 0; u < 4; ++u)
      acc = dot8_bf16(((const bf16x8 *)xl)[c + 16 * u], wv[u], acc);
  }
  for (; c < nc; c += 16)
    acc = dot8_bf16(((const bf16x8 *)xl)[c], ((const bf16x8 *)wr)[c], acc);

#pragma unroll
  for (int off = 8; off > 0; off >>= 1) acc += __shfl_xor(acc, off, WAVE);
  if (sl == 0 && n0 < N) y[n0] = f32_to_bf16(acc * rms);
}

// residual-add epilogue variant: resid[n] += x @ W[n] (in place).
extern "C" __global__ void __launch_bounds__(256)
gemv_res_w32_kernel(const ushort_t *__restrict__ x,
                    const ushort_t *__restrict__ w,
                    ushort_t *__restrict__ resid, int K, int N) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int rg = lane >> 5;
  const int sl = lane & 31;
  const int n = blockIdx.x * 8 + wid * 2 + rg;
  if (n >= N) return;

  const ushort_t *wr = w + (size_t)n * K;
  const int nc = K / 8;
  // prefetch the residual word now: a dependent load AFTER the dot loop
  // adds its full latency to the kernel tail (probe: +0.25 us/launch)
  const float r0 = bf16_to_f32(resid[n]);

  float acc = 0.f;
  int c = sl;
  for (; c + 224 < nc; c += 256) {
    bf16x8 wv[8], xv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      wv[u] = ((const bf16x8 *)wr)[c + 32 * u];
      xv[u] = ((const bf16x8 *)x)[c + 32 * u];
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) acc = dot8_bf16(xv[u], wv[u], acc);
  }
  for (; c < nc; c += 32)
    acc = dot8_bf16(((const bf16x8 *)x)[c], ((const bf16x8 *)wr)[c], acc);

#pragma unroll
  for (int off = 16; off > 0; off >>= 1) acc += __shfl_xor(acc, off, WAVE);
  if (sl == 0) resid[n] = f32_to_bf16(r0 + acc);
}

extern "C" __global__ void __launch_bounds__(256)
gemv_gateup_norm_kernel(const ushort_t *__restrict__ x,
                        const ushort_t *__restrict__ wln,
                        const ushort_t *__restrict__ w,
                        ushort_t *__restrict__ act, int K, int F, float eps) {
  extern __shared__ ushort_t xl[];
  __shared__ float red[16];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int rg = lane >> 5;
  const int sl = lane & 31;
  const int n0 = blockIdx.x * 8 + wid * 2 + rg;
  const int n = n0 < F ? n0 : F - 1;

  const float rms = norm_stage_lds(x, wln, xl, red, K, eps);

  const ushort_t *wg = w + (size_t)n * K;
  const ushort_t *wu = w + (size_t)(n + F) * K;
  const int nc = K / 8;

  float ag = 0.f, au = 0.f;
  int c = sl;
  for (; c + 96 < nc; c += 128) {
    bf16x8 gv[4], uv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      gv[u] = ((const bf16x8 *)wg)[c + 32 * u];
      uv[u] = ((const bf16x8 *)wu)[c + 32 * u];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const bf16x8 xv = ((const bf16x8 *)xl)[c + 32 * u];
      ag = dot8_bf16(xv, gv[u], ag);
      au = dot8_bf16(xv, uv[u], au);
    }
  }
  for (; c < nc; c += 32) {
    const bf16x8 xv = ((const bf16x8 *)xl)[c];
    ag = dot8_bf16(xv, ((const bf16x8 *)wg)[c], ag);
    au = dot8_bf16(xv, ((const bf16x8 *)wu)[c], au);
  }

#pragma unroll
  for (int off = 16; off > 0; off >>= 1) {
    ag += __shfl_xor(ag, off, WAVE);
    au += __shfl_xor(au, off, WAVE);
  }
  if (sl == 0 && n0 < F) {
    const float g = ag * rms;
    const float s = g / (1.0f + __expf(-g));  // silu(gate)
    act[n0] = f32_to_bf16(s * (au * rms));
  }
}

extern "C" void launch_gemv_norm(const ushort_t *x, const ushort_t *wln,
                                 const ushort_t *w, ushort_t *y, int K, int N,
                                 float eps, hipStream_t stream) {
  const size_t lds = (size_t)K * 2;  // xl staging (x*wln as bf16)
  if (N <= 8192)
    gemv_norm_w32_kernel<<<dim3((N + 7) / 8), 256, lds, stream>>>(x, wln, w,
                                                                  y, K, N,
                                                                  eps);
  else
    gemv_norm_kernel<<<dim3((N + 15) / 16), 256, lds, stream>>>(x, wln, w, y,
                                                                K, N, eps);
}

extern "C" void launch_gemv_res(const ushort_t *x, const ushort_t *w,
                                ushort_t *resid, int K, int N,
                                hipStream_t stream) {
  gemv_res_w32_kernel<<<dim3((N + 7) / 8), 256, 0, stream>>>(x, w, resid, K, N);
}

extern "C" void launch_gemv_gateup_norm(const ushort_t *x, const ushort_t *wln,
                                        const ushort_t *w, ushort_t *act,
                                        int K, int F, float eps,
                                        hipStream_t stream) {
  gemv_gateup_norm_kernel<<<dim3((F + 7) / 8), 256, (size_t)K * 2, stream>>>(
      x, wln, w, act, K, F, eps);
}
