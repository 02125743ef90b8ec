// Flash-style causal prefill attention on CDNA4 MFMA (gfx950), bf16 I/O,
// fp32 accumulation. This is the hot prefill kernel for hd=128 models
// (Llama-3 family); hd∈{32,64} falls back to attention.hip's simple kernel.
//
// Structure (cdna_hip_programming.md §5/§B):
//   grid  = (hq, ceil(tq/128));  block = 512 threads (8 waves)
//   each wave owns a 16-row Q tile; the block shares LDS-staged K/V tiles
//   of KVBLK=64 keys (64 halves the softmax passes + syncs vs 32).
//
// LDS images (all carved from ONE array — compiler trap §5 4a):
//   K  [64][128] bf16, XOR-swizzled byte^=((key&7)<<4): a 256-B row puts a
//      column read's whole 16-lane group on one bank without it
//      (Guideline 4's D=128 hazard — 16-way, measured 5.1 ms/layer).
//   Vt [128][VPITCH=72] bf16: V TRANSPOSED with 64-key rows padded to 72
//      elements (row stride 144 B) plus a per-row rotation of the key index
//      (krow+rot)&63, so the PV B-fragment is one ds_read_b128 per lane
//      with the 16 consecutive-row banks distinct, not 8 scalar reads.
//   P  [16][72] bf16 per wave (C->A layout bounce), same 144-B row pitch.
//
// MFMA fragment maps (hardware-verified by tests/test_ops_gpu.py mfma probe):
//   mfma_f32_16x16x32_bf16:
//     A: lane l holds A[row = l%16][k = (l/16)*8 + j], j = 0..7
//     B: lane l holds B[k = (l/16)*8 + j][col = l%16]
//     C/D: lane l, reg r holds C[row = (l>>4)*4 + r][col = l&15]

#include "common.h"

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8v;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4v;

#define QROWS 16   // q rows per wave
#define KVBLK 64   // keys per staged tile (64: halves softmax passes+syncs)
#define NWAVE 8    // waves per block
#define HD 128     // head dim (this kernel is hd=128 only)
#define VPITCH 72  // padded row length (elements) of the Vt and P images

// ABL: perf-ablation variants (guide §5 mistake #8 — ablate before
// optimizing; rule 17 keeps stubbed values live via asm). 0 = full
// kernel (the production path); 1 = no softmax (raw scores feed PV);
// 2 = no P LDS bounce (reuse a fixed pfrag); 3 = no PV MFMAs.
template <int ABL>
__global__ void __launch_bounds__(512, 1)
attn_prefill_mfma_kernel(const ushort_t *__restrict__ q,
                         const ushort_t *__restrict__ k,
                         const ushort_t *__restrict__ v,
                         ushort_t *__restrict__ out, int tq, int tk,
                         int kv_offset, float scale, int hq, int kh,
                         int causal) {
  const int h = blockIdx.x;
  const int g = h / (hq / kh);   // kv head
  const int qblock = blockIdx.y; // 128 q rows per block
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;

  const int qb0 = qblock * (NWAVE * QROWS);
  const int q0 = qb0 + wid * QROWS; // this wave's first q row

  __shared__ __attribute__((aligned(16))) ushort_t lds[
      KVBLK * HD + HD * VPITCH + NWAVE * QROWS * VPITCH];
  ushort_t *ldsK = lds;                       // swizzled [64][128]
  ushort_t *ldsVt = lds + KVBLK * HD;         // [128][72]
  ushort_t *ldsP = ldsVt + HD * VPITCH + wid * QROWS * VPITCH;

  const int lrow = lane & 15;       // 0..15
  const int lhi = lane >> 4;        // 0..3

  // ---- load Q fragments (persistent): 4 k-chunks of 32 dims ----
  bf16x8v qfrag[4];
  {
    const int qrow = q0 + lrow;
    const bool ok = qrow < tq;
    const ushort_t *qr = q + ((size_t)(ok ? qrow : 0) * hq + h) * HD;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      if (ok) {
        qfrag[c] = ((const bf16x8v *)qr)[c * 4 + lhi];  // dims c*32+lhi*8..
      } else {
        qfrag[c] = (bf16x8v){0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  }

  // ---- state ----
  // SWAPPED-QK^T layout (guide T12 idea): S^T = mfma(K, Q) puts the
  // whole softmax row of THIS LANE's q-row (= lrow) in registers, so the
  // online-softmax state is one scalar pair per lane and the row reduce
  // is an in-lane tree + 2 shfls (the row-major form needed 8 chains of
  // 4 shfls per tile — the ablation showed softmax at 43% of the kernel).
  float m1 = -INFINITY, l1 = 0.f;  // row = lrow
  f32x4v o[8];  // O accumulators: 8 dim-chunks of 16 (rows lhi*4+r)
#pragma unroll
  for (int d = 0; d < 8; ++d) o[d] = (f32x4v){0.f, 0.f, 0.f, 0.f};

  int kmax = tk;
  if (causal) kmax = min(tk, kv_offset + qb0 + NWAVE * QROWS);

  // ---- async-STAGE split (guide T14, +17% on this ladder): the next
  // tile's K/V global loads are ISSUED right after the barrier that frees
  // the LDS image, and the register->LDS write happens one iteration
  // later — the ~full-tile compute phase (32 MFMAs + softmax) hides the
  // HBM latency that the old load-then-store staging exposed serially.
  bf16x8 kst[2], vst[2];
  auto stage_load = [&](int kt) {
#pragma unroll
    for (int piece = 0; piece < 2; ++piece) {
      const int flat = tid + 512 * piece;
      const int krow = flat >> 4;       // key within tile: 0..63
      const int kcol8 = flat & 15;      // 16B chunk within the 256B row
      const int key = kt + krow;
      if (key < tk) {
        kst[piece] = ((const bf16x8 *)(k + ((size_t)key * kh + g) * HD))[kcol8];
        vst[piece] = ((const bf16x8 *)(v + ((size_t)key * kh + g) * HD))[kcol8];
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) { kst[piece].u[j] = 0; vst[piece].u[j] = 0; }
      }
    }
  };
  auto stage_write = [&]() {
#pragma unroll
    for (int piece = 0; piece < 2; ++piece) {
      const int flat = tid + 512 * piece;
      const int krow = flat >> 4;
      const int kcol8 = flat & 15;
      // K: swizzled 16B store
      unsigned kbyte = (unsigned)flat * 16u;
      kbyte ^= ((unsigned)(krow & 7)) << 4;
      *(bf16x8 *)((char *)ldsK + kbyte) = kst[piece];
      // Vt: 8 scalar transposed stores with the KEY SLOT rotated by a
      // dim-derived amount (raw pattern = one bank for a whole write
      // group; rotation spreads it 8-wide with static indexing; reads
      // de-rotate and each 16 B vector stays in an 8-aligned 64-slot
      // window, no wrap).
      const int d0 = kcol8 * 8;
      const int rot = (kcol8 & 7) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        ldsVt[(size_t)(d0 + j) * VPITCH + ((krow + rot) & 63)] = vst[piece].u[j];
      }
    }
  };

  stage_load(0);
  for (int kt = 0; kt < kmax; kt += KVBLK) {
    __syncthreads();  // previous iteration's reads done
    stage_write();
    __syncthreads();
    if (kt + KVBLK < kmax) stage_load(kt + KVBLK);  // hide under compute

    // ---- S^T = K Q^T for four 16-key subtiles (swapped operands: the
    // 16x16x32 A and B fragment layouts coincide, so the SAME fragments
    // serve; C becomes S^T[key][q-row] with the row lane-local) ----
    f32x4v s[4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      s[n] = (f32x4v){0.f, 0.f, 0.f, 0.f};
      const int keyr = n * 16 + lrow;
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        // A frag: A[key][kd] = K[keyr][c*32+kd8] from the swizzled image
        unsigned kbyte = (unsigned)keyr * 256u + (unsigned)(c * 4 + lhi) * 16u;
        kbyte ^= ((unsigned)(keyr & 7)) << 4;
        const bf16x8v afr = *(const bf16x8v *)((const char *)ldsK + kbyte);
        s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afr, qfrag[c], s[n], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    if (ABL >= 1) {  // skip softmax; keep scores live (rule 17)
#pragma unroll
      for (int n = 0; n < 4; ++n)
        asm volatile("" :: "v"(s[n][0]), "v"(s[n][1]), "v"(s[n][2]),
                     "v"(s[n][3]));
      if (ABL < 3) {
        if (ABL < 2) {
#pragma unroll
          for (int n = 0; n < 4; ++n)
#pragma unroll
            for (int r = 0; r < 4; ++r)
              ldsP[(lhi * 4 + r) * VPITCH + n * 16 + lrow] =
                  f32_to_bf16(s[n][r]);
        }
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int half = 0; half < 2; ++half) {
          const bf16x8v pfrag =
              *(const bf16x8v *)(ldsP + lrow * VPITCH + half * 32 + lhi * 8);
#pragma unroll
          for (int d = 0; d < 8; ++d) {
            const int vdim = d * 16 + lrow;
            const int vrot = (((unsigned)vdim >> 3) & 7) * 8;
            const bf16x8v vfr = *(const bf16x8v *)(
                ldsVt + (size_t)vdim * VPITCH +
                ((half * 32 + lhi * 8 + vrot) & 63));
            o[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfr, o[d],
                                                           0, 0, 0);
          }
        }
        __builtin_amdgcn_s_setprio(0);
      } else {
#pragma unroll
        for (int d = 0; d < 8; ++d)
          asm volatile("" :: "v"(o[d][0]), "v"(o[d][3]));
      }
      continue;
    }

    // ---- online softmax, lane-local row (row = lrow) ----
    // Interior tiles of a causal 8k+ walk are FULLY unmasked for the
    // whole wave (uniform predicate): skip the 32 mask compares there.
    const bool full_tile =
        (kt + KVBLK <= tk) &&
        (!causal || kt + KVBLK <= kv_offset + q0 + 1);
    // softmax runs in the LOG2-SCALED domain (exp2+fma fold, ladder
    // item): raw scores stay unscaled; m/l track m2 = m*scale*log2e and
    // p = exp2(fma(s_raw, scale2, -m2)) — one v_fma + v_exp per element
    // instead of mul + sub + (mul+exp). Masked keys set s_raw = -inf
    // (the fma propagates it).
    if (!full_tile) {
      const int qrow_abs = kv_offset + q0 + lrow;
#pragma unroll
      for (int n = 0; n < 4; ++n) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key = kt + n * 16 + lhi * 4 + r;
          if (key >= tk || (causal && key > qrow_abs))
            s[n][r] = -INFINITY;
        }
      }
    }
    const float scale2 = scale * 1.4426950408889634f;
    // in-lane 16-value max tree + 2 cross-lane levels (lanes lrow,
    // lrow+16, lrow+32, lrow+48 hold the rest of this row's keys)
    float mx = fmaxf(fmaxf(s[0][0], s[0][1]), fmaxf(s[0][2], s[0][3]));
#pragma unroll
    for (int n = 1; n < 4; ++n)
      mx = fmaxf(mx, fmaxf(fmaxf(s[n][0], s[n][1]),
                           fmaxf(s[n][2], s[n][3])));
    mx = fmaxf(mx, __shfl_xor(mx, 16, WAVE));
    mx = fmaxf(mx, __shfl_xor(mx, 32, WAVE));
    const float mx2 = mx * scale2;  // log2-scaled row max (one op)
    // defer-max (guide T13): when no row's max grew by more than THR=8
    // (= 8*log2e in this domain), keep the old running max — P is then
    // bounded by e^8 (fine in f32 accum and scale-free bf16) and the
    // O-rescale pass is skipped (+5% measured on this ladder; ~3x
    // max-abs error per T13's numbers, inside the kernel tests'
    // tolerances). Wave-uniform (__all); first-tile m1 = -inf forces
    // the rescale branch. Order is T13-safe: each tile's PV completes
    // before the next tile's decision, and l sees the same alpha.
    const bool defer = __all(mx2 - m1 <= 11.5416f);
    const float mn = defer ? m1 : fmaxf(m1, mx2);
    const float alpha =
        defer ? 1.0f : ((mn == -INFINITY) ? 0.f : exp2f(m1 - mn));
    float psum = 0.f;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = exp2f(__builtin_fmaf(s[n][r], scale2, -mn));
        s[n][r] = p;
        psum += p;
      }
    }
    psum += __shfl_xor(psum, 16, WAVE);
    psum += __shfl_xor(psum, 32, WAVE);
    l1 = l1 * alpha + psum;
    m1 = mn;
    if (!defer) {
      // O rows are lhi*4+r: fetch those rows' alphas (any lane with
      // lrow == row has the value; pick the same lhi quartile)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float oa = __shfl(alpha, (lhi * 4 + r) | (lhi << 4), WAVE);
#pragma unroll
        for (int d = 0; d < 8; ++d) o[d][r] *= oa;
      }
    }

    // ---- P -> LDS: lane-local row, 4 keys packed per 8-B store ----
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      union { uint32_t u32[2]; ushort_t u16[4]; } pk;
#pragma unroll
      for (int r = 0; r < 4; ++r) pk.u16[r] = f32_to_bf16(s[n][r]);
      *(uint2 *)(ldsP + (size_t)lrow * VPITCH + n * 16 + lhi * 4) =
          make_uint2(pk.u32[0], pk.u32[1]);
    }

    // ---- O += P V (two 32-key halves) ----
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const bf16x8v pfrag =
          *(const bf16x8v *)(ldsP + lrow * VPITCH + half * 32 + lhi * 8);
#pragma unroll
      for (int d = 0; d < 8; ++d) {
        // B frag: B[kd][col] = V[kd8][d*16+col] = Vt[d*16+col][kd8] — one
        // contiguous 16B LDS read per lane.
        const int vdim = d * 16 + lrow;
        const int vrot = (((unsigned)vdim >> 3) & 7) * 8;
        const bf16x8v vfr = *(const bf16x8v *)(
            ldsVt + (size_t)vdim * VPITCH +
            ((half * 32 + lhi * 8 + vrot) & 63));
        o[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfr, o[d], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // ---- epilogue: O / l (l lives on the lane whose lrow == row) ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + lhi * 4 + r;
    if (qrow >= tq) continue;
    const float lr = __shfl(l1, (lhi * 4 + r) | (lhi << 4), WAVE);
    const float inv = (lr > 0.f) ? 1.0f / lr : 0.f;
    ushort_t *orow = out + ((size_t)qrow * hq + h) * HD;
#pragma unroll
    for (int d = 0; d < 8; ++d) {
      orow[d * 16 + lrow] = f32_to_bf16(o[d][r] * inv);
    }
  }
}

extern "C" void launch_attn_prefill_mfma(const ushort_t *q, const ushort_t *k,
                                         const ushort_t *v, ushort_t *out,
                                         int tq, int tk, int kv_offset,
                                         float scale, int hq, int kh,
                                         int hd, hipStream_t stream, int *ok) {
  if (hd != HD) {
    *ok = 0;
    return;
  }
  dim3 grid(hq, (tq + NWAVE * QROWS - 1) / (NWAVE * QROWS));
  attn_prefill_mfma_kernel<0><<<grid, NWAVE * WAVE, 0, stream>>>(
      q, k, v, out, tq, tk, kv_offset, scale, hq, kh, 1);
  *ok = 1;
}

// perf-ablation entry (tools/attn_ablate probe; never on the hot path)
extern "C" void launch_attn_prefill_mfma_abl(
    const ushort_t *q, const ushort_t *k, const ushort_t *v, ushort_t *out,
    int tq, int tk, float scale, int hq, int kh, int abl,
    hipStream_t stream) {
  dim3 grid(hq, (tq + NWAVE * QROWS - 1) / (NWAVE * QROWS));
#define ABL_CASE(N)                                                           \
  case N:                                                                     \
    attn_prefill_mfma_kernel<N><<<grid, NWAVE * WAVE, 0, stream>>>(           \
        q, k, v, out, tq, tk, 0, scale, hq, kh, 1);                           \
    break;
  switch (abl) {
    ABL_CASE(0)
    ABL_CASE(1)
    ABL_CASE(2)
    ABL_CASE(3)
  }
#undef ABL_CASE
}

// ---------------------------------------------------------------------------
// MFMA layout probe: D = A @ B for A[16,32], B[32,16] bf16 — used by the GPU
// test suite to pin the fragment maps against torch.matmul before trusting
// the attention kernel.
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(64)
mfma_probe_16x16x32(const ushort_t *__restrict__ a,  // [16][32]
                    const ushort_t *__restrict__ b,  // [32][16]
                    float *__restrict__ d) {         // [16][16]
  const int lane = threadIdx.x & (WAVE - 1);
  const int lrow = lane & 15, lhi = lane >> 4;
  bf16x8v af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = (short)a[lrow * 32 + lhi * 8 + j];
    bf[j] = (short)b[(lhi * 8 + j) * 16 + lrow];
  }
  f32x4v c = (f32x4v){0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) d[(lhi * 4 + r) * 16 + lrow] = c[r];
}
