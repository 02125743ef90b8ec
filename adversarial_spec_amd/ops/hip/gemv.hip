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

// one 16 B chunk of the wln-weighted dot + sum(x^2), f32 math. (The
// two-kernel sequence rounds the normed activation to bf16 between the
// kernels; the fused form skips that round — slightly MORE accurate.)
DEVINL void dot8_norm(const bf16x8 &x, const bf16x8 &l, const bf16x8 &w,
                      float &acc, float &s2) {
  const f32x8 xf = unpack8(x), lf = unpack8(l), wf = unpack8(w);
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    acc = fmaf(xf.v[i] * lf.v[i], wf.v[i], acc);
    s2 = fmaf(xf.v[i], xf.v[i], s2);
  }
}

extern "C" __global__ void __launch_bounds__(256)
gemv_norm_w32_kernel(const ushort_t *__restrict__ x,
                     const ushort_t *__restrict__ wln,
                     const ushort_t *__restrict__ w,
                     ushort_t *__restrict__ y, int K, int N, float eps) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int rg = lane >> 5;
  const int sl = lane & 31;
  const int n = blockIdx.x * 8 + wid * 2 + rg;
  if (n >= N) return;

  const ushort_t *wr = w + (size_t)n * K;
  const int nc = K / 8;

  float acc = 0.f, s2 = 0.f;
  int c = sl;
  // 4-deep (not 8): three streams in flight (w + L2-resident x, wln)
  for (; c + 96 < nc; c += 128) {
    bf16x8 wv[4], xv[4], lv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      wv[u] = ((const bf16x8 *)wr)[c + 32 * u];
      xv[u] = ((const bf16x8 *)x)[c + 32 * u];
      lv[u] = ((const bf16x8 *)wln)[c + 32 * u];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) dot8_norm(xv[u], lv[u], wv[u], acc, s2);
  }
  for (; c < nc; c += 32)
    dot8_norm(((const bf16x8 *)x)[c], ((const bf16x8 *)wln)[c],
              ((const bf16x8 *)wr)[c], acc, s2);

#pragma unroll
  for (int off = 16; off > 0; off >>= 1) {
    acc += __shfl_xor(acc, off, WAVE);
    s2 += __shfl_xor(s2, off, WAVE);
  }
  if (sl == 0) y[n] = f32_to_bf16(acc * rsqrtf(s2 / K + eps));
}

// 16-lane-group norm variant for LARGE N (lm_head: final_norm fused)
extern "C" __global__ void __launch_bounds__(256)
gemv_norm_kernel(const ushort_t *__restrict__ x,
                 const ushort_t *__restrict__ wln,
                 const ushort_t *__restrict__ w,
                 ushort_t *__restrict__ y, int K, int N, float eps) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int rg = lane >> 4;
  const int sl = lane & 15;
  const int n = blockIdx.x * 16 + wid * 4 + rg;
  if (n >= N) return;

  const ushort_t *wr = w + (size_t)n * K;
  const int nc = K / 8;

  float acc = 0.f, s2 = 0.f;
  int c = sl;
  for (; c + 48 < nc; c += 64) {
    bf16x8 wv[4], xv[4], lv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      wv[u] = ((const bf16x8 *)wr)[c + 16 * u];
      xv[u] = ((const bf16x8 *)x)[c + 16 * u];
      lv[u] = ((const bf16x8 *)wln)[c + 16 * u];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) dot8_norm(xv[u], lv[u], wv[u], acc, s2);
  }
  for (; c < nc; c += 16)
    dot8_norm(((const bf16x8 *)x)[c], ((const bf16x8 *)wln)[c],
              ((const bf16x8 *)wr)[c], acc, s2);

#pragma unroll
  for (int off = 8; off > 0; off >>= 1) {
    acc += __shfl_xor(acc, off, WAVE);
    s2 += __shfl_xor(s2, off, WAVE);
  }
  if (sl == 0) y[n] = f32_to_bf16(acc * rsqrtf(s2 / K + eps));
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
  if (sl == 0) resid[n] = f32_to_bf16(bf16_to_f32(resid[n]) + acc);
}

// one 16 B chunk for the norm-fused gate_up pair
DEVINL void dot8_norm2(const bf16x8 &x, const bf16x8 &l, const bf16x8 &wg,
                       const bf16x8 &wu, float &ag, float &au, float &s2) {
  const f32x8 xf = unpack8(x), lf = unpack8(l);
  const f32x8 gf = unpack8(wg), uf = unpack8(wu);
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const float xl = xf.v[i] * lf.v[i];
    ag = fmaf(xl, gf.v[i], ag);
    au = fmaf(xl, uf.v[i], au);
    s2 = fmaf(xf.v[i], xf.v[i], s2);
  }
}

extern "C" __global__ void __launch_bounds__(256)
gemv_gateup_norm_kernel(const ushort_t *__restrict__ x,
                        const ushort_t *__restrict__ wln,
                        const ushort_t *__restrict__ w,
                        ushort_t *__restrict__ act, int K, int F, float eps) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int rg = lane >> 5;
  const int sl = lane & 31;
  const int n = blockIdx.x * 8 + wid * 2 + rg;
  if (n >= F) return;

  const ushort_t *wg = w + (size_t)n * K;
  const ushort_t *wu = w + (size_t)(n + F) * K;
  const int nc = K / 8;

  float ag = 0.f, au = 0.f, s2 = 0.f;
  int c = sl;
  for (; c + 96 < nc; c += 128) {
    bf16x8 gv[4], uv[4], xv[4], lv[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      gv[u] = ((const bf16x8 *)wg)[c + 32 * u];
      uv[u] = ((const bf16x8 *)wu)[c + 32 * u];
      xv[u] = ((const bf16x8 *)x)[c + 32 * u];
      lv[u] = ((const bf16x8 *)wln)[c + 32 * u];
    }
#pragma unroll
    for (int u = 0; u < 4; ++u)
      dot8_norm2(xv[u], lv[u], gv[u], uv[u], ag, au, s2);
  }
  for (; c < nc; c += 32)
    dot8_norm2(((const bf16x8 *)x)[c], ((const bf16x8 *)wln)[c],
               ((const bf16x8 *)wg)[c], ((const bf16x8 *)wu)[c], ag, au, s2);

#pragma unroll
  for (int off = 16; off > 0; off >>= 1) {
    ag += __shfl_xor(ag, off, WAVE);
    au += __shfl_xor(au, off, WAVE);
    s2 += __shfl_xor(s2, off, WAVE);
  }
  if (sl == 0) {
    const float rms = rsqrtf(s2 / K + eps);
    const float g = ag * rms;
    const float s = g / (1.0f + __expf(-g));  // silu(gate)
    act[n] = f32_to_bf16(s * (au * rms));
  }
}

extern "C" void launch_gemv_norm(const ushort_t *x, const ushort_t *wln,
                                 const ushort_t *w, ushort_t *y, int K, int N,
                                 float eps, hipStream_t stream) {
  if (N <= 8192)
    gemv_norm_w32_kernel<<<dim3((N + 7) / 8), 256, 0, stream>>>(x, wln, w, y,
                                                                K, N, eps);
  else
    gemv_norm_kernel<<<dim3((N + 15) / 16), 256, 0, stream>>>(x, wln, w, y,
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
  gemv_gateup_norm_kernel<<<dim3((F + 7) / 8), 256, 0, stream>>>(x, wln, w,
                                                                 act, K, F,
                                                                 eps);
}
