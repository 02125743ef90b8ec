// Tiled MFMA GEMM for CDNA4 (gfx950): C[M,N] = A[M,K] @ B[N,K]^T,
// bf16 in/out, fp32 accumulation via v_mfma_f32_16x16x32_bf16.
//
// Weights are stored ROW-MAJOR [out, in] (= HF checkpoint layout, no
// load-time transpose), so the A and B tiles stage IDENTICALLY: both are
// [128 rows][64 k] images whose 16 B fragment reads are contiguous
// ds_read_b128 — no LDS transpose pass anywhere.
//
// This replaces library GEMMs (hipBLASLt/rocBLAS) on the prefill path:
// besides the perf headroom, Tensile's skinny-m bf16 kernels return
// garbage intermittently in recycled-memory states (round-1 debugging).
// In-tree, deterministic, workspace-free, graph-capture safe.
//
// Structure = the cdna_hip_programming.md "step 3" ladder
// (128x128 tile, BK=64, double-buffered `global_load_lds` staging,
// one barrier per K-step, ~900 TF/s at 4096^3 on this chip):
//   - block 256 threads = 4 waves in a 2x2 quadrant grid, each wave owns
//     a 64x64 output quadrant = 4x4 MFMA 16x16 tiles;
//   - staging via __builtin_amdgcn_global_load_lds width 16: LDS stays
//     lane-linear, the bank-conflict XOR swizzle is applied to the
//     PER-LANE GLOBAL SOURCE address (chunk ^= row&7 within each 128 B
//     row — stays inside the cacheline, so coalescing is preserved);
//   - fragment reads de-swizzle with the same XOR.
// Edge tiles (M/N/K not tile-aligned) take a guarded register-staging
// path into the same swizzled image, so the MFMA loop is shared.

#include "common.h"

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8v;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4v;

#define GT_M 128
#define GT_N 128
#define GT_K 64  // elements; one row of the LDS image = 128 B

DEVINL void glds16(const ushort_t *g, ushort_t *l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void *)g,
      (__attribute__((address_space(3))) void *)l, 16, 0, 0);
}

// Stage one [128][64] bf16 tile via glds. Each wave-instruction covers
// 1 KiB = 8 rows x 128 B; 4 waves x 4 issues cover the 16 KiB image.
// src row r, 16B-chunk c is fetched from global chunk (c ^ (r & 7)) so the
// lane-linear LDS placement realizes the swizzled image.
DEVINL void stage_glds(const ushort_t *__restrict__ src, long row_stride,
                       ushort_t *__restrict__ img, int tid) {
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int r0 = (wid * 4 + i) * 8;
    const int row = r0 + (lane >> 3);
    const int c = lane & 7;
    const ushort_t *g = src + (size_t)row * row_stride + (size_t)(c ^ (row & 7)) * 8;
    // LDS dest: wave-uniform base + lane*16 (lane-linear within the 1 KiB)
    glds16(g, img + (size_t)r0 * GT_K);
  }
}

// Guarded register staging for edge tiles: same swizzled image layout.
// 1024 16 B chunk-pieces (128 rows x 8 chunks), 4 per thread.
DEVINL void stage_edge(const ushort_t *__restrict__ src, long row_stride,
                       int rows_left, int k_left, ushort_t *__restrict__ img,
                       int tid) {
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int piece = tid + 256 * p;   // 0..1023
    const int row = piece >> 3;        // 0..127
    const int c = piece & 7;           // 16B chunk
    bf16x8 v;
    const int gk = c * 8;
    if (row < rows_left && gk < k_left) {
      const ushort_t *g = src + (size_t)row * row_stride + gk;
      if (gk + 8 <= k_left) {
        v = *(const bf16x8 *)g;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) v.u[j] = (gk + j < k_left) ? g[j] : 0;
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) v.u[j] = 0;
    }
    ((bf16x8 *)(img + (size_t)row * GT_K))[c ^ (row & 7)] = v;
  }
}

__global__ void __launch_bounds__(256, 2)
gemm_nt_kernel(const ushort_t *__restrict__ a, const ushort_t *__restrict__ b,
               ushort_t *__restrict__ c_out, int M, int N, int K) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;
  const int lrow = lane & 15;
  const int lhi = lane >> 4;

  const int n0 = blockIdx.x * GT_N;
  const int m0 = blockIdx.y * GT_M;
  const int wm = (wid >> 1) * 64;  // wave quadrant
  const int wn = (wid & 1) * 64;

  __shared__ __attribute__((aligned(16))) ushort_t imgA[2][GT_M * GT_K];
  __shared__ __attribute__((aligned(16))) ushort_t imgB[2][GT_N * GT_K];

  f32x4v acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4v){0.f, 0.f, 0.f, 0.f};

  const int ntiles = (K + GT_K - 1) / GT_K;
  // block-uniform edge predicates: only edge blocks (and the last K tile
  // when K % 64 != 0) pay the guarded register-staging path
  const bool a_edge = (M - m0) < GT_M;
  const bool b_edge = (N - n0) < GT_N;

  auto stage = [&](int kt, int buf) {
    const ushort_t *asrc = a + (size_t)m0 * K + (size_t)kt * GT_K;
    const ushort_t *bsrc = b + (size_t)n0 * K + (size_t)kt * GT_K;
    const bool k_edge = (kt + 1) * GT_K > K;
    if (a_edge || k_edge) {
      stage_edge(asrc, K, M - m0, K - kt * GT_K, imgA[buf], tid);
    } else {
      stage_glds(asrc, K, imgA[buf], tid);
    }
    if (b_edge || k_edge) {
      stage_edge(bsrc, K, N - n0, K - kt * GT_K, imgB[buf], tid);
    } else {
      stage_glds(bsrc, K, imgB[buf], tid);
    }
  };

  stage(0, 0);
  __syncthreads();  // drains the glds queue (vmcnt0 inside the barrier)

  for (int kt = 0; kt < ntiles; ++kt) {
    const int buf = kt & 1;
    if (kt + 1 < ntiles) stage(kt + 1, buf ^ 1);  // prefetch next tile

    // ---- 2 k-subtiles of 32: 8 frag reads + 16 MFMAs each ----
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      bf16x8v afr[4], bfr[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = wm + i * 16 + lrow;
        const int ch = (s * 4 + lhi) ^ (row & 7);
        afr[i] = ((const bf16x8v *)(imgA[buf] + (size_t)row * GT_K))[ch];
      }
#pragma unroll
      for (int j2 = 0; j2 < 4; ++j2) {
        const int row = wn + j2 * 16 + lrow;
        const int ch = (s * 4 + lhi) ^ (row & 7);
        bfr[j2] = ((const bf16x8v *)(imgB[buf] + (size_t)row * GT_K))[ch];
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j2 = 0; j2 < 4; ++j2)
          acc[i][j2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i], bfr[j2], acc[i][j2], 0, 0, 0);
    }
    __syncthreads();  // compute done + prefetch glds drained
  }

  // ---- store C: D frag lane l -> col l%16, rows (l/16)*4 + 0..3 ----
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j2 = 0; j2 < 4; ++j2) {
      const int col = n0 + wn + j2 * 16 + lrow;
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm + i * 16 + lhi * 4 + r;
        if (row >= M) continue;
        c_out[(size_t)row * N + col] = f32_to_bf16(acc[i][j2][r]);
      }
    }
  }
}

extern "C" void launch_gemm(const ushort_t *a, const ushort_t *b, ushort_t *c,
                            int M, int N, int K, hipStream_t stream) {
  dim3 grid((N + GT_N - 1) / GT_N, (M + GT_M - 1) / GT_M);
  gemm_nt_kernel<<<grid, 256, 0, stream>>>(a, b, c, M, N, K);
}
