// Common device helpers for the adversarial_spec_amd CDNA4 (gfx950) kernels.
//
// Conventions (cdna_hip_programming.md):
//   - wave width is 64 (hard-coded, not warpSize-derived)
//   - bf16 memory traffic is vectorized as ushort4/ushort8 (8-16 B/lane)
//   - accumulation is fp32
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

typedef unsigned short ushort_t;

// ---- bf16 <-> f32 -------------------------------------------------------

DEVINL float bf16_to_f32(ushort_t u) {
  union { uint32_t i; float f; } c;
  c.i = ((uint32_t)u) << 16;
  return c.f;
}

DEVINL ushort_t f32_to_bf16(float f) {
  union { uint32_t i; float f; } c;
  c.f = f;
  // round-to-nearest-even
  uint32_t lsb = (c.i >> 16) & 1u;
  c.i += 0x7fffu + lsb;
  return (ushort_t)(c.i >> 16);
}

// 8 bf16 packed in 16 bytes (one dwordx4 load)
struct bf16x8 { ushort_t u[8]; };
struct f32x8  { float v[8]; };

DEVINL f32x8 unpack8(const bf16x8 &p) {
  f32x8 r;
#pragma unroll
  for (int i = 0; i < 8; ++i) r.v[i] = bf16_to_f32(p.u[i]);
  return r;
}

DEVINL bf16x8 pack8(const f32x8 &p) {
  bf16x8 r;
#pragma unroll
  for (int i = 0; i < 8; ++i) r.u[i] = f32_to_bf16(p.v[i]);
  return r;
}

// packed-bf16 dot: acc += sum_j a[j]*b[j] over one 16 B chunk (8 elems)
// via 4 v_dot2_f32_bf16 — 4 VALU ops instead of 24 (16 unpacks + 8 FMAs).
typedef __attribute__((__vector_size__(2 * sizeof(short)))) short bf16x2v_;

DEVINL float dot8_bf16(const bf16x8 &a, const bf16x8 &b, float acc) {
  const bf16x2v_ *ap = (const bf16x2v_ *)&a;
  const bf16x2v_ *bp = (const bf16x2v_ *)&b;
#pragma unroll
  for (int q = 0; q < 4; ++q)
    acc = __builtin_amdgcn_fdot2_f32_bf16(ap[q], bp[q], acc, false);
  return acc;
}

// ---- wave/block reductions ---------------------------------------------

DEVINL float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

DEVINL float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block reduction over up to 16 waves via LDS; every thread returns the result.
// `scratch` must hold >= 16 floats. blockDim.x must be a multiple of 64.
DEVINL float block_reduce_sum(float v, float *scratch) {
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;
  v = wave_reduce_sum(v);
  if ((threadIdx.x & (WAVE - 1)) == 0) scratch[wid] = v;
  __syncthreads();
  float r = 0.f;
#pragma unroll 16
  for (int i = 0; i < nw; ++i) r += scratch[i];
  __syncthreads();
  return r;
}

DEVINL float block_reduce_max(float v, float *scratch) {
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;
  v = wave_reduce_max(v);
  if ((threadIdx.x & (WAVE - 1)) == 0) scratch[wid] = v;
  __syncthreads();
  float r = -INFINITY;
#pragma unroll 16
  for (int i = 0; i < nw; ++i) r = fmaxf(r, scratch[i]);
  __syncthreads();
  return r;
}

// ---- RNG (counter-based, for the sampling kernel) -----------------------

DEVINL uint32_t hash_u32(uint32_t x) {
  x ^= x >> 16; x *= 0x7feb352du;
  x ^= x >> 15; x *= 0x846ca68bu;
  x ^= x >> 16;
  return x;
}

DEVINL float uniform01(uint32_t seed) {
  // (hash in (0,1]); never exactly 0 so log() etc. stay safe
  return (hash_u32(seed) + 1u) * (1.0f / 4294967296.0f);
}

#define HIP_CHECK_LAST()                                                       \
  do {                                                                         \
    hipError_t _e = hipGetLastError();                                         \
    if (_e != hipSuccess) {                                                    \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(_e)); \
    }                                                                          \
  } while (0)
