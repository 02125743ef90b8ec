// fp8 (OCP e4m3) variant of the 256x256 8-phase deep-pipelined GEMM:
//   C[M,N] = (A8[M,K] * asc[M]) @ (B8[N,K] * bsc[N])^T, fp32 accumulation,
// on v_mfma_f32_16x16x32_fp8_fp8 (rowwise scales factor out of the dot
// product — exact dequantization in the epilogue).
//
// Same schedule as gemm256.hip (see its header for the invariant): one
// half-tile staged per phase 6 phases ahead, counted vmcnt at the K-tile
// boundary, raw barriers, setprio'd 16-MFMA quadrant clusters. fp8
// halves every byte count: a half-image is [128][64] fp8 = 8 KiB, one
// glds piece per wave per half-tile, so the steady-state wait is
// vmcnt(2) (two half-tiles in flight).
//
// LDS swizzle: 16-B chunks XOR'd with ((row >> 2) & 3) within the 64-B
// row. NOT (row & 3): 64-B rows repeat a 256-B bank row every 4 rows, so
// rows {r, r+4, r+8, r+12} of a 16-lane read group alias the same banks
// at the same intra-row chunk — the low-bit XOR (which the 2-barrier
// 128^2 fp8 kernel gets away with, its stage stall hiding LDS reads)
// leaves that 4-way conflict in place, and in the 8-phase schedule the
// LDS read IS the critical path (guide §5.5 T2 regime gate). XORing with
// row bits 2-3 gives the four aliasing rows four different chunks.
// Fragments are 8-B reads of the low/high half of a swizzled chunk.

#include "common.h"

typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4v;

#define Q8_BM 256
#define Q8_BN 256
#define Q8_BK 64
#define Q8_IMG (128 * 64)  // BYTES per half-image

DEVINL void q8_glds16(const uint8_t *g, uint8_t *l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void *)g,
      (__attribute__((address_space(3))) void *)l, 16, 0, 0);
}

// Stage half-image H of K-tile u: ONE 1-KiB glds piece per wave
// (8 waves x 16 rows x 64 B = the full 8 KiB image).
template <int H>
DEVINL void q8_stage(const uint8_t *const sbase[4][2], int u,
                     uint8_t *__restrict__ lds, int wid) {
  uint8_t *img = lds + (size_t)((u & 1) * 4 + H) * Q8_IMG;
  const long kb = (long)u * Q8_BK;
  q8_glds16(sbase[H][0] + kb, img + (size_t)wid * 16 * Q8_BK);
}

template <int NSTAGE, int WAITK>
DEVINL void q8_tile(const uint8_t *__restrict__ imA,
                    const uint8_t *__restrict__ imB, int t,
                    const uint8_t *const sbase[4][2],
                    uint8_t *__restrict__ lds, int wid, int lrow, int lhi,
                    int brow0, long (&afr)[8][2], long (&bfr)[4][2],
                    f32x4v (&acc)[8][4]) {
  // fragment read: 8 fp8 at k-slice ks*32 + lhi*8 of the 64-B row,
  // chunk-swizzled (chunk ^ (row & 3)), low/high half by lhi parity
  auto frag = [&](const uint8_t *im, int row, int ks) -> long {
    const int ch = ((ks * 2) + (lhi >> 1)) ^ ((row >> 2) & 3);
    return *(const long *)(im + (size_t)row * Q8_BK + ch * 16 +
                           (lhi & 1) * 8);
  };

  // ---------- phase 0
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    const int row = mi * 16 + lrow;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) afr[mi][ks] = frag(imA, row, ks);
  }
#pragma unroll
  for (int ni = 0; ni < 2; ++ni) {
    const int row = brow0 + ni * 16 + lrow;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) bfr[ni][ks] = frag(imB, row, ks);
  }
  if (NSTAGE > 0) q8_stage<2>(sbase, t + 1, lds, wid);
  __builtin_amdgcn_s_barrier();
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
  __builtin_amdgcn_s_setprio(1);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            afr[mi][ks], bfr[ni][ks], acc[mi][ni], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  __builtin_amdgcn_s_barrier();

  // ---------- phase 1
#pragma unroll
  for (int mi = 4; mi < 8; ++mi) {
    const int row = mi * 16 + lrow;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) afr[mi][ks] = frag(imA, row, ks);
  }
#pragma unroll
  for (int ni = 2; ni < 4; ++ni) {
    const int row = brow0 + (ni - 2) * 16 + 32 + lrow;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) bfr[ni][ks] = frag(imB, row, ks);
  }
  if (NSTAGE > 1) q8_stage<3>(sbase, t + 1, lds, wid);
  __builtin_amdgcn_s_barrier();
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
  __builtin_amdgcn_s_setprio(1);
#pragma unroll
  for (int mi = 4; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            afr[mi][ks], bfr[ni][ks], acc[mi][ni], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  __builtin_amdgcn_s_barrier();

  // ---------- phase 2
  if (NSTAGE > 2) q8_stage<0>(sbase, t + 2, lds, wid);
  __builtin_amdgcn_s_barrier();
  __builtin_amdgcn_s_setprio(1);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 2; ni < 4; ++ni)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            afr[mi][ks], bfr[ni][ks], acc[mi][ni], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  __builtin_amdgcn_s_barrier();

  // ---------- phase 3 + boundary wait
  if (NSTAGE > 3) q8_stage<1>(sbase, t + 2, lds, wid);
  __builtin_amdgcn_s_barrier();
  __builtin_amdgcn_s_setprio(1);
#pragma unroll
  for (int mi = 4; mi < 8; ++mi)
#pragma unroll
    for (int ni = 2; ni < 4; ++ni)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            afr[mi][ks], bfr[ni][ks], acc[mi][ni], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  if (WAITK == 2) {
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  } else if (WAITK == 0) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();
}

__global__ void __launch_bounds__(512, 2)
gemm_fp8_256_kernel(const uint8_t *__restrict__ a,
                    const float *__restrict__ asc,
                    const uint8_t *__restrict__ b,
                    const float *__restrict__ bsc,
                    ushort_t *__restrict__ c_out, int M, int N, int K) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int lrow = lane & 15;
  const int lhi = lane >> 4;
  const int wm = wid >> 2;
  const int wn = wid & 3;

  const int n0 = blockIdx.x * Q8_BN;
  const int m0 = blockIdx.y * Q8_BM;
  const int nt = K / Q8_BK;

  __shared__ __attribute__((aligned(16))) uint8_t lds[2 * 4 * Q8_IMG];

  // per-lane stage source pointers at k=0 (row clamped, chunk swizzled):
  // piece rows = wid*16 + lane/4, chunk = lane%4
  const uint8_t *sbase[4][2];
#pragma unroll
  for (int h = 0; h < 4; ++h) {
    const uint8_t *src = (h < 2) ? a : b;
    const int row0 = (h < 2) ? (m0 + h * 128) : (n0 + (h - 2) * 128);
    const int rmax = ((h < 2) ? M : N) - 1;
    const int rl = wid * 16 + (lane >> 2);
    const int grow = min(row0 + rl, rmax);
    const int c = lane & 3;
    sbase[h][0] = src + (size_t)grow * K + (size_t)(c ^ ((rl >> 2) & 3)) * 16;
    sbase[h][1] = sbase[h][0];  // single piece per wave (layout parity)
  }

  f32x4v acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4v){0.f, 0.f, 0.f, 0.f};

  q8_stage<0>(sbase, 0, lds, wid);
  q8_stage<1>(sbase, 0, lds, wid);
  q8_stage<2>(sbase, 0, lds, wid);
  q8_stage<3>(sbase, 0, lds, wid);
  q8_stage<0>(sbase, 1, lds, wid);
  q8_stage<1>(sbase, 1, lds, wid);
  asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  const size_t offA = (size_t)wm * Q8_IMG;
  const size_t offB = (size_t)(2 + (wn >> 1)) * Q8_IMG;
  const int brow0 = (wn & 1) * 64;

  long afr[8][2];
  long bfr[4][2];

  int t = 0;
  for (; t + 2 < nt; ++t) {
    const uint8_t *imA = lds + (size_t)(t & 1) * 4 * Q8_IMG + offA;
    const uint8_t *imB = lds + (size_t)(t & 1) * 4 * Q8_IMG + offB;
    q8_tile<4, 2>(imA, imB, t, sbase, lds, wid, lrow, lhi, brow0, afr, bfr,
                  acc);
  }
  {
    const uint8_t *imA = lds + (size_t)(t & 1) * 4 * Q8_IMG + offA;
    const uint8_t *imB = lds + (size_t)(t & 1) * 4 * Q8_IMG + offB;
    q8_tile<2, 0>(imA, imB, t, sbase, lds, wid, lrow, lhi, brow0, afr, bfr,
                  acc);
    ++t;
  }
  {
    const uint8_t *imA = lds + (size_t)(t & 1) * 4 * Q8_IMG + offA;
    const uint8_t *imB = lds + (size_t)(t & 1) * 4 * Q8_IMG + offB;
    q8_tile<0, -1>(imA, imB, t, sbase, lds, wid, lrow, lhi, brow0, afr, bfr,
                   acc);
  }

  // epilogue with rowwise dequant scales
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int col = n0 + wn * 64 + ni * 16 + lrow;
      if (col >= N) continue;
      const float bs = bsc[col];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm * 128 + mi * 16 + lhi * 4 + r;
        if (row >= M) continue;
        c_out[(size_t)row * N + col] =
            f32_to_bf16(acc[mi][ni][r] * asc[row] * bs);
      }
    }
  }
}

extern "C" void launch_gemm_fp8_256(const uint8_t *a, const float *asc,
                                    const uint8_t *b, const float *bsc,
                                    ushort_t *c, int M, int N, int K,
                                    hipStream_t stream) {
  dim3 grid((N + Q8_BN - 1) / Q8_BN, (M + Q8_BM - 1) / Q8_BM);
  gemm_fp8_256_kernel<<<grid, 512, 0, stream>>>(a, asc, b, bsc, c, M, N, K);
}
