// Deep-pipelined 256x256 MFMA GEMM for CDNA4 (gfx950):
// C[M,N] = A[M,K] @ B[N,K]^T, bf16 in/out, fp32 accumulation.
//
// This is the 8-phase 256^2 structure from cdna_hip_programming.md §5
// ("The 256^2 8-phase template"): the 128^2 two-barrier kernel (gemm.hip)
// ceilings at ~900 TF/s because the workgroup-release inside its
// __syncthreads drains the global_load_lds queue (vmcnt(0)) every K-step.
// Here the staging loads SPAN barriers behind a counted s_waitcnt:
//
//   - tile 256x256, BK=64, 8 waves (2M x 4N), one block per CU;
//   - LDS = 128 KiB: 2 buffers x 4 half-images [128][64] bf16
//     (A rows 0-127, A rows 128-255, B rows 0-127, B rows 128-255);
//   - 4 phases per K-tile; each phase: {ds_read a fragment subtile,
//     glds-prefetch ONE half-tile of a future K-tile, raw s_barrier,
//     lgkmcnt(0), setprio(1), 16 MFMA (one C-quadrant x K=64),
//     setprio(0), raw s_barrier};
//   - half-tiles are staged 6 phases ahead of first use; the ONLY vmcnt
//     in the main loop is a counted vmcnt(4) once per K-tile (two
//     half-tiles = 4 glds/wave stay in flight across every barrier);
//     the last boundary drains (vmcnt 0) and the epilogue stores.
//
// Schedule invariant (why vmcnt(4) is correct): stages issue one
// half-tile per phase in order S = 4*tile + half (phase 0 of tile t
// stages (t+1, B_lo), phase 1 (t+1, B_hi), phase 2 (t+2, A_lo), phase 3
// (t+2, A_hi)); the wait at the end of tile t's phase 3 must cover every
// half of tile t+1. (t+1, h3) is issued at phase 4t+1, and exactly two
// stages — (t+2, h0), (t+2, h1) — follow it before the wait, so "all but
// the newest 4 loads" == "tile t+1 fully landed". Image overwrite is
// safe because a buffer's images are last READ in phases 0-1 of their
// tile (all fragment ds_reads happen there) and first REWRITTEN 4+
// phases later. The tail tiles are peeled as template specializations so
// the main loop carries NO stage guards: a guarded variant spilled 3
// VGPRs and hipcc's spill-reload wait (vmcnt(0)) drained the pipeline
// every iteration.
//
// Swizzle: identical to gemm.hip — glds writes lane-linear, so the
// bank-conflict XOR (chunk ^= row&7) is applied to the per-lane GLOBAL
// source address (within one 128-B cacheline, coalescing preserved) and
// de-swizzled on the fragment ds_read (guide §5.4 rule 21).
//
// Edge handling: M/N edges CLAMP the per-lane source row (reads stay in
// bounds; garbage rows only feed C rows/cols that the guarded epilogue
// never stores). K must be a multiple of 128 and >= 512 — the dispatcher
// (bindings.hip gemm()) falls back to gemm.hip otherwise.

#include "common.h"

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8v;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4v;

#define G2_BM 256
#define G2_BN 256
#define G2_BK 64
#define G2_IMG (128 * 64)  // elements per half-image

DEVINL void g2_glds16(const ushort_t *g, ushort_t *l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void *)g,
      (__attribute__((address_space(3))) void *)l, 16, 0, 0);
}

// Stage half-image H of K-tile u. sbase[H][i] is this lane's precomputed
// source pointer for k=0 (row-clamped, chunk-swizzled); each wave issues
// 2 glds of 1 KiB (8 rows x 128 B each).
template <int H>
DEVINL void g2_stage(const ushort_t *const sbase[4][2], int u,
                     ushort_t *__restrict__ lds, int wid) {
  ushort_t *img = lds + (size_t)((u & 1) * 4 + H) * G2_IMG;
  const long kb = (long)u * G2_BK;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    g2_glds16(sbase[H][i] + kb, img + (size_t)(wid * 2 + i) * 8 * G2_BK);
  }
}

// One K-tile: 4 phases. NSTAGE phases issue a half-tile prefetch
// (phase q stages half number 4t+6+q); WAITK = counted vmcnt immediate at
// the tile boundary (-1: none).
template <int NSTAGE, int WAITK>
DEVINL void g2_tile(const ushort_t *__restrict__ imA,
                    const ushort_t *__restrict__ imB, int t,
                    const ushort_t *const sbase[4][2],
                    ushort_t *__restrict__ lds, int wid, int lrow, int lhi,
                    int brow0, bf16x8v (&afr)[8][2], bf16x8v (&bfr)[4][2],
                    f32x4v (&acc)[8][4]) {
  // ---------- phase 0: read A rows 0-63 + B rows 0-63 (12 ds_read_b128),
  //            stage (t+1, B_lo), MFMA quadrant (mi 0-3, ni 0-1)
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    const int row = mi * 16 + lrow;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
      afr[mi][ks] = ((const bf16x8v *)(imA + (size_t)row * G2_BK))
          [(ks * 4 + lhi) ^ (row & 7)];
  }
#pragma unroll
  for (int ni = 0; ni < 2; ++ni) {
    const int row = brow0 + ni * 16 + lrow;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
      bfr[ni][ks] = ((const bf16x8v *)(imB + (size_t)row * G2_BK))
          [(ks * 4 + lhi) ^ (row & 7)];
  }
  if (NSTAGE > 0) g2_stage<2>(sbase, t + 1, lds, wid);
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
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afr[mi][ks], bfr[ni][ks], acc[mi][ni], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  __builtin_amdgcn_s_barrier();

  // ---------- phase 1: read A rows 64-127 (8 reads),
  //            stage (t+1, B_hi), MFMA quadrant (mi 4-7, ni 0-1)
#pragma unroll
  for (int mi = 4; mi < 8; ++mi) {
    const int row = mi * 16 + lrow;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
      afr[mi][ks] = ((const bf16x8v *)(imA + (size_t)row * G2_BK))
          [(ks * 4 + lhi) ^ (row & 7)];
  }
  if (NSTAGE > 1) g2_stage<3>(sbase, t + 1, lds, wid);
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
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afr[mi][ks], bfr[ni][ks], acc[mi][ni], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  __builtin_amdgcn_s_barrier();

  // ---------- phase 2: read B rows 64-127 (4 reads);
  //            stage (t+2, A_lo); MFMA (mi 0-3, ni 2-3)
#pragma unroll
  for (int ni = 2; ni < 4; ++ni) {
    const int row = brow0 + (ni - 2) * 16 + 32 + lrow;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
      bfr[ni][ks] = ((const bf16x8v *)(imB + (size_t)row * G2_BK))
          [(ks * 4 + lhi) ^ (row & 7)];
  }
  if (NSTAGE > 2) g2_stage<0>(sbase, t + 2, lds, wid);
  __builtin_amdgcn_s_barrier();
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
  __builtin_amdgcn_s_setprio(1);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 2; ni < 4; ++ni)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afr[mi][ks], bfr[ni][ks], acc[mi][ni], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  __builtin_amdgcn_s_barrier();

  // ---------- phase 3: no reads; stage (t+2, A_hi); MFMA (mi 4-7, ni 2-3);
  //            counted K-tile boundary wait
  if (NSTAGE > 3) g2_stage<1>(sbase, t + 2, lds, wid);
  __builtin_amdgcn_s_barrier();
  __builtin_amdgcn_s_setprio(1);
#pragma unroll
  for (int mi = 4; mi < 8; ++mi)
#pragma unroll
    for (int ni = 2; ni < 4; ++ni)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afr[mi][ks], bfr[ni][ks], acc[mi][ni], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  if (WAITK == 4) {
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else if (WAITK == 0) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();
}

__global__ void __launch_bounds__(512, 2)
gemm_nt256_kernel(const ushort_t *__restrict__ a,
                  const ushort_t *__restrict__ b,
                  ushort_t *__restrict__ c_out, int M, int N, int K) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;   // 0..7
  const int lrow = lane & 15;
  const int lhi = lane >> 4;
  const int wm = wid >> 2;      // wave M half: rows wm*128 .. +127
  const int wn = wid & 3;       // wave N quarter: cols wn*64 .. +63

  const int n0 = blockIdx.x * G2_BN;
  const int m0 = blockIdx.y * G2_BM;
  const int nt = K / G2_BK;     // K % 128 == 0, K >= 512 (dispatcher)

  // ONE shared array (a second __shared__ object de-pipelines glds —
  // guide §5 ".s-level traps" (a)).
  __shared__ __attribute__((aligned(16))) ushort_t lds[2 * 4 * G2_IMG];

  // Per-lane stage source pointers at k=0, one per (half-image, issue):
  // row clamped into bounds (edge blocks read valid garbage whose C rows
  // are never stored), chunk index XOR-swizzled within the 128-B row.
  const ushort_t *sbase[4][2];
#pragma unroll
  for (int h = 0; h < 4; ++h) {
    const ushort_t *src = (h < 2) ? a : b;
    const int row0 = (h < 2) ? (m0 + h * 128) : (n0 + (h - 2) * 128);
    const int rmax = ((h < 2) ? M : N) - 1;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int rl = (wid * 2 + i) * 8 + (lane >> 3);  // 0..127 in image
      const int grow = min(row0 + rl, rmax);
      const int c = lane & 7;
      sbase[h][i] = src + (size_t)grow * K + (size_t)(c ^ (rl & 7)) * 8;
    }
  }

  f32x4v acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4v){0.f, 0.f, 0.f, 0.f};

  // ---- prologue: stage tile 0 + halves A_lo/A_hi of tile 1; wait tile 0
  g2_stage<0>(sbase, 0, lds, wid);
  g2_stage<1>(sbase, 0, lds, wid);
  g2_stage<2>(sbase, 0, lds, wid);
  g2_stage<3>(sbase, 0, lds, wid);
  g2_stage<0>(sbase, 1, lds, wid);
  g2_stage<1>(sbase, 1, lds, wid);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  // per-wave image base offsets (within a buffer)
  const size_t offA = (size_t)wm * G2_IMG;            // image h0 or h1
  const size_t offB = (size_t)(2 + (wn >> 1)) * G2_IMG;
  const int brow0 = (wn & 1) * 64;                    // rows in B image

  bf16x8v afr[8][2];
  bf16x8v bfr[4][2];

  int t = 0;
  for (; t + 2 < nt; ++t) {
    const ushort_t *imA = lds + (size_t)(t & 1) * 4 * G2_IMG + offA;
    const ushort_t *imB = lds + (size_t)(t & 1) * 4 * G2_IMG + offB;
    g2_tile<4, 4>(imA, imB, t, sbase, lds, wid, lrow, lhi, brow0, afr, bfr,
                  acc);
  }
  {  // tile nt-2: only B_lo/B_hi of tile nt-1 remain to stage; full drain
    const ushort_t *imA = lds + (size_t)(t & 1) * 4 * G2_IMG + offA;
    const ushort_t *imB = lds + (size_t)(t & 1) * 4 * G2_IMG + offB;
    g2_tile<2, 0>(imA, imB, t, sbase, lds, wid, lrow, lhi, brow0, afr, bfr,
                  acc);
    ++t;
  }
  {  // tile nt-1: nothing in flight, nothing to stage
    const ushort_t *imA = lds + (size_t)(t & 1) * 4 * G2_IMG + offA;
    const ushort_t *imB = lds + (size_t)(t & 1) * 4 * G2_IMG + offB;
    g2_tile<0, -1>(imA, imB, t, sbase, lds, wid, lrow, lhi, brow0, afr, bfr,
                   acc);
  }

  // ---- epilogue: D frag lane l -> col l%16, rows (l/16)*4 + 0..3
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int col = n0 + wn * 64 + ni * 16 + lrow;
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm * 128 + mi * 16 + lhi * 4 + r;
        if (row >= M) continue;
        c_out[(size_t)row * N + col] = f32_to_bf16(acc[mi][ni][r]);
      }
    }
  }
}

extern "C" void launch_gemm256(const ushort_t *a, const ushort_t *b,
                               ushort_t *c, int M, int N, int K,
                               hipStream_t stream) {
  dim3 grid((N + G2_BN - 1) / G2_BN, (M + G2_BM - 1) / G2_BM);
  gemm_nt256_kernel<<<grid, 512, 0, stream>>>(a, b, c, M, N, K);
}
