// Fused memory-bound kernels: RMSNorm, residual-add+RMSNorm, RoPE, SwiGLU,
// paged KV scatter. All bf16 I/O with fp32 accumulation, vectorized 8 bf16
// (16 B) per lane per access (cdna_hip_programming.md Guideline 13).
//
// MI355X-first design notes:
//  - one workgroup per row for the norms (t rows >> 256 CUs in prefill);
//  - the fused add_rmsnorm writes the residual stream once (HBM-bound op:
//    3 reads + 2 writes fused vs 4 reads + 3 writes unfused);
//  - RoPE reads host-precomputed f32 cos/sin tables (no device trig);
//  - all kernels are grid-stride free (static shapes from the caller).

#include "common.h"

// ---------------------------------------------------------------------------
// RMSNorm: y[r,:] = x[r,:] / rms * w.  One 256-thread block per row.
// d must be a multiple of 8. Each thread covers d/256/8 vectors (or strided).
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(1024)
rmsnorm_kernel(const ushort_t *__restrict__ x, const ushort_t *__restrict__ w,
               ushort_t *__restrict__ y, int d, float eps) {
  __shared__ float scratch[16];
  const int row = blockIdx.x;
  const ushort_t *xr = x + (size_t)row * d;
  ushort_t *yr = y + (size_t)row * d;
  const int nvec = d / 8;

  float ss = 0.f;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    f32x8 v = unpack8(((const bf16x8 *)xr)[i]);
#pragma unroll
    for (int j = 0; j < 8; ++j) ss += v.v[j] * v.v[j];
  }
  ss = block_reduce_sum(ss, scratch);
  const float inv = rsqrtf(ss / d + eps);

  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    f32x8 v = unpack8(((const bf16x8 *)xr)[i]);
    f32x8 ww = unpack8(((const bf16x8 *)w)[i]);
    f32x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o.v[j] = v.v[j] * inv * ww.v[j];
    ((bf16x8 *)yr)[i] = pack8(o);
  }
}

// ---------------------------------------------------------------------------
// Fused residual add + RMSNorm:
//   r' = resid + delta   (written back)
//   y  = rmsnorm(r') * w
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(1024)
add_rmsnorm_kernel(const ushort_t *__restrict__ resid,
                   const ushort_t *__restrict__ delta,
                   const ushort_t *__restrict__ w,
                   ushort_t *__restrict__ resid_out,
                   ushort_t *__restrict__ y, int d, float eps) {
  __shared__ float scratch[16];
  const int row = blockIdx.x;
  const size_t base = (size_t)row * d;
  const int nvec = d / 8;

  // pass 1: add + sum of squares; keep r' in bf16 (the stored precision) so
  // the written residual and the normalized output agree bit-for-bit with
  // a separate add-then-norm.
  float ss = 0.f;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    f32x8 a = unpack8(((const bf16x8 *)(resid + base))[i]);
    f32x8 b = unpack8(((const bf16x8 *)(delta + base))[i]);
    f32x8 r;
#pragma unroll
    for (int j = 0; j < 8; ++j) r.v[j] = a.v[j] + b.v[j];
    bf16x8 rp = pack8(r);
    ((bf16x8 *)(resid_out + base))[i] = rp;
    f32x8 rq = unpack8(rp);
#pragma unroll
    for (int j = 0; j < 8; ++j) ss += rq.v[j] * rq.v[j];
  }
  ss = block_reduce_sum(ss, scratch);
  const float inv = rsqrtf(ss / d + eps);

  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    f32x8 r = unpack8(((const bf16x8 *)(resid_out + base))[i]);
    f32x8 ww = unpack8(((const bf16x8 *)w)[i]);
    f32x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o.v[j] = r.v[j] * inv * ww.v[j];
    ((bf16x8 *)(y + base))[i] = pack8(o);
  }
}

// ---------------------------------------------------------------------------
// RoPE in place, interleaved-pair convention.
//   q: [t, hq, hd]  k: [t, hk, hd]   cos/sin: [max_seq, hd/2] f32
// One wave per (token, head); lane L rotates pair (2L, 2L+1) when hd==128
// (hd/2 == 64 pairs == one lane each). For hd < 128, lanes beyond hd/2 idle;
// for hd > 128 lanes loop.
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(256)
rope_kernel(ushort_t *__restrict__ q, ushort_t *__restrict__ k,
            const float *__restrict__ cost, const float *__restrict__ sint,
            int t, int hq, int hk, int hd, int pos0,
            long q_rstride, long k_rstride,
            const int *__restrict__ pos_ptr) {
  // HIP-graph decode: the position lives in a device word so one captured
  // graph replays for every token (pos_ptr == nullptr -> use pos0).
  if (pos_ptr) pos0 = *pos_ptr;
  // q/k may be strided views into the fused QKV GEMM output (row strides in
  // elements); head/dim dims are contiguous.
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int heads = hq + hk;
  const int tok = wave_id / heads;
  if (tok >= t) return;
  const int head = wave_id % heads;
  const int pos = pos0 + tok;
  const int half = hd / 2;

  ushort_t *base = (head < hq)
                       ? q + (size_t)tok * q_rstride + (size_t)head * hd
                       : k + (size_t)tok * k_rstride + (size_t)(head - hq) * hd;
  const float *crow = cost + (size_t)pos * half;
  const float *srow = sint + (size_t)pos * half;

  for (int p = lane; p < half; p += WAVE) {
    uint32_t pair = ((uint32_t *)base)[p];  // 2 bf16: (even, odd)
    float ev = bf16_to_f32((ushort_t)(pair & 0xffffu));
    float od = bf16_to_f32((ushort_t)(pair >> 16));
    float c = crow[p], s = srow[p];
    float e2 = ev * c - od * s;
    float o2 = ev * s + od * c;
    ((uint32_t *)base)[p] =
        (uint32_t)f32_to_bf16(e2) | ((uint32_t)f32_to_bf16(o2) << 16);
  }
}

// ---------------------------------------------------------------------------
// Fused RoPE + paged KV scatter. Decode is kernel-count bound (~450
// launches/token); this replaces the rope_kernel + kv_write_kernel pair and
// also saves one full re-read of K (rope read+wrote it, kv_write re-read it).
//   waves map over t * (hq + 2*hk) "slots":
//     slot < hq:            rotate q in place
//     hq <= slot < hq+hk:   rotate k in place AND scatter the rotated row
//                           into the cache page pool
//     slot >= hq+hk:        copy v into the cache page pool
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(256)
rope_kv_kernel(ushort_t *__restrict__ q, ushort_t *__restrict__ k,
               const ushort_t *__restrict__ v,
               const float *__restrict__ cost, const float *__restrict__ sint,
               ushort_t *__restrict__ kc, ushort_t *__restrict__ vc,
               const int *__restrict__ page_table, int t, int hq, int hk,
               int hd, int pos0, long q_rstride, long k_rstride,
               long v_rstride, int page, const int *__restrict__ pos_ptr) {
  if (pos_ptr) pos0 = *pos_ptr;
  const int wave_id = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int slots = hq + 2 * hk;
  const int tok = wave_id / slots;
  if (tok >= t) return;
  const int slot = wave_id % slots;
  const int pos = pos0 + tok;
  const int half = hd / 2;

  const int phys = page_table[pos / page];
  const size_t cache_row = ((size_t)phys * page + (pos % page)) * hk * hd;

  if (slot >= hq + hk) {  // v copy: no rotation
    const int vh = slot - hq - hk;
    const bf16x8 *src =
        (const bf16x8 *)(v + (size_t)tok * v_rstride + (size_t)vh * hd);
    bf16x8 *dst = (bf16x8 *)(vc + cache_row + (size_t)vh * hd);
    for (int i = lane; i < hd / 8; i += WAVE) dst[i] = src[i];
    return;
  }

  const bool is_q = slot < hq;
  ushort_t *base = is_q
                       ? q + (size_t)tok * q_rstride + (size_t)slot * hd
                       : k + (size_t)tok * k_rstride + (size_t)(slot - hq) * hd;
  uint32_t *cache_dst =
      is_q ? nullptr
           : (uint32_t *)(kc + cache_row + (size_t)(slot - hq) * hd);
  const float *crow = cost + (size_t)pos * half;
  const float *srow = sint + (size_t)pos * half;

  for (int p = lane; p < half; p += WAVE) {
    uint32_t pair = ((uint32_t *)base)[p];  // 2 bf16: (even, odd)
    float ev = bf16_to_f32((ushort_t)(pair & 0xffffu));
    float od = bf16_to_f32((ushort_t)(pair >> 16));
    float c = crow[p], s = srow[p];
    const uint32_t rot = (uint32_t)f32_to_bf16(ev * c - od * s) |
                         ((uint32_t)f32_to_bf16(ev * s + od * c) << 16);
    ((uint32_t *)base)[p] = rot;
    if (cache_dst) cache_dst[p] = rot;
  }
}

// ---------------------------------------------------------------------------
// SwiGLU: out = silu(gate) * up, elementwise over [n] (n = t * ffn).
// gate/up are the two contiguous halves of the fused gate_up GEMM output,
// passed as separate base pointers with a row stride.
//   gate,up: [t, f] views with row stride `stride` elements.
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(256)
swiglu_kernel(const ushort_t *__restrict__ gate, const ushort_t *__restrict__ up,
              ushort_t *__restrict__ out, int t, int f, int in_stride) {
  const size_t nvec = (size_t)t * (f / 8);
  const int fv = f / 8;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < nvec; i += (size_t)gridDim.x * blockDim.x) {
    const size_t row = i / fv, col = i % fv;
    const size_t in_off = row * (in_stride / 8) + col;
    f32x8 g = unpack8(((const bf16x8 *)gate)[in_off]);
    f32x8 u = unpack8(((const bf16x8 *)up)[in_off]);
    f32x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gv = g.v[j];
      const float sig = 1.0f / (1.0f + __expf(-gv));
      o.v[j] = gv * sig * u.v[j];
    }
    ((bf16x8 *)out)[row * fv + col] = pack8(o);
  }
}

// ---------------------------------------------------------------------------
// Paged KV scatter: write t fresh K/V rows into the page pool.
//   k,v:        [t, kh, hd] bf16 (contiguous)
//   kc,vc:      [n_pages, page, kh, hd] bf16
//   page_table: int32 [n_logical_pages]
// One block per token; threads copy kh*hd elements vectorized.
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(256)
kv_write_kernel(const ushort_t *__restrict__ k, const ushort_t *__restrict__ v,
                ushort_t *__restrict__ kc, ushort_t *__restrict__ vc,
                const int *__restrict__ page_table, int pos0, int t,
                int kh, int hd, int page,
                const int *__restrict__ pos_ptr) {
  if (pos_ptr) pos0 = *pos_ptr;
  const int tok = blockIdx.x;
  if (tok >= t) return;
  const int pos = pos0 + tok;
  const int phys = page_table[pos / page];
  const size_t row = (size_t)phys * page + (pos % page);
  const int n = kh * hd / 8;
  const bf16x8 *ks = (const bf16x8 *)(k + (size_t)tok * kh * hd);
  const bf16x8 *vs = (const bf16x8 *)(v + (size_t)tok * kh * hd);
  bf16x8 *kd = (bf16x8 *)(kc + row * kh * hd);
  bf16x8 *vd = (bf16x8 *)(vc + row * kh * hd);
  for (int i = threadIdx.x; i < n; i += blockDim.x) {
    kd[i] = ks[i];
    vd[i] = vs[i];
  }
}
