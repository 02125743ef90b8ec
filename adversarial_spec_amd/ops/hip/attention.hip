// Attention kernels for CDNA4 (gfx950).
//
//  - attn_decode_split / attn_decode_combine: single-token decode over the
//    paged KV cache, flash-decoding style (split-KV online softmax with a
//    combine pass). Memory-bound by the KV read: each KV row is read ONCE
//    per kv-head and serves all `group` GQA query heads; vectorized 16 B
//    per lane; splits spread the read over the whole chip.
//
//  - attn_prefill_simple: correctness-first causal prefill over the fresh
//    contiguous K/V of the prompt (one 16-lane group per query row, online
//    softmax in registers). It is the numerics anchor; the MFMA-tiled
//    prefill (attn_prefill_mfma.hip) replaces it on the hot path.
//
// Layout contracts (ops/torch_ref.py):
//   q        [tq, hq, hd] bf16
//   k,v      [tk, kh, hd] bf16 (contiguous, prefill)
//   kc,vc    [n_pages, page, kh, hd] bf16 (decode)
//   page_table int32
// State is fp32 throughout; hd in {32, 64, 128}; group = hq/kh <= 8.

#include "common.h"

#define MAXG 8

// 16-lane-group dot reduce (LPP = lanes per position = hd/8)
template <int LPP>
DEVINL float group_reduce_sum(float v) {
#pragma unroll
  for (int off = LPP / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// ---------------------------------------------------------------------------
// Decode split kernel.
// grid.x = kh, grid.y = n_splits, block = 256 (4 waves).
//
// The KV read is staged through LDS by async global_load_lds in 64-position
// double-buffered tiles (LDS-DMA keeps a whole tile in flight behind a
// counted vmcnt — register-ring prefetch measured latency-bound).
//
// Compute is GROUP-PER-WAVE, LANE-PER-POSITION: wave w owns GQA head
// group w (and w+4 at group 8); in the QK pass lane p holds position
// tbase+p's full dot (the K image is chunk-XOR-swizzled so 16-lane b128
// groups reading the same chunk index of 16 different rows stay
// conflict-free), so the online-softmax reduce is ONE 6-level wave
// reduction per 64-position tile — the old dim-sliced layout paid 4
// shfl-chain levels per position per group (64+ chained DS ops per
// tile). P values reach the dim-sliced PV pass by shfl broadcast: no
// ds_write anywhere in the kernel (an LDS store makes hipcc order every
// stage-image ds_read behind vmcnt(0), draining the DMA pipeline —
// .s-verified round-2 failure mode).
//
// Workspace (fp32): ws_m, ws_l: [kh, n_splits, group]
//                   ws_acc:     [kh, n_splits, group, hd]
// ---------------------------------------------------------------------------

#define DTILE 64   // positions staged per LDS tile

template <int LPP, int MG, bool IDENT>
__global__ void __launch_bounds__(256) attn_decode_split_kernel(
    const ushort_t *__restrict__ q,   // [hq, hd]
    const ushort_t *__restrict__ kc, const ushort_t *__restrict__ vc,
    const int *__restrict__ page_table, int seq_len, float scale,
    int kh, int group, int hd, int page, int split_len,
    float *__restrict__ ws_m, float *__restrict__ ws_l,
    float *__restrict__ ws_acc, const int *__restrict__ pos_ptr,
    ushort_t *__restrict__ out) {
  // graph mode: seq_len = *pos_ptr + 1 (attend up to and incl. the token
  // kv_write just appended at position *pos_ptr)
  if (pos_ptr) seq_len = *pos_ptr + 1;
  const int g = blockIdx.x;          // kv head
  const int split = blockIdx.y;
  const int n_splits = gridDim.y;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int sub = lane / LPP;        // PV pass: position sub-group
  const int sl = lane % LPP;         // PV pass: dim slice within position
  const int subs = WAVE / LPP;       // concurrent positions per PV step
  constexpr int NG = (MG + 3) / 4;   // GQA heads owned per wave

  const int start = split * split_len;
  const int limit = min(seq_len, (split + 1) * split_len);

  // ONE shared array (guide §5 trap 4a): [2 stage buffers][K | V] + the
  // q image. K images are chunk-swizzled; V and q are linear.
  constexpr int TILE_E = DTILE * (LPP * 8);          // elements per tile
  constexpr int QIMG_E = MG * LPP * 8;
  __shared__ __attribute__((aligned(16))) ushort_t lds[4 * TILE_E + QIMG_E];
  ushort_t *qlds = lds + 4 * TILE_E;

  // K chunk swizzle: a 256-B LDS bank row holds 32/LPP image rows, so
  // same-chunk reads of two rows collide whenever the rows share a
  // bank-row offset (p ≡ p' mod 16/LPP); XOR the chunk index with the
  // row's bank-alias class to spread them.
  constexpr int SWZ_DIV = 16 / LPP;  // rows sharing a bank-row offset
  auto kswz = [&](int row) { return (row / SWZ_DIV) & (LPP - 1); };

  float m[NG], l[NG], acc[NG][8];
#pragma unroll
  for (int gi = 0; gi < NG; ++gi) {
    m[gi] = -INFINITY;
    l[gi] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[gi][j] = 0.f;
  }

  // stage one DTILE-position K/V tile into LDS buffer b via glds.
  // Each wave issues LPP/4 x 1 KiB pieces for K and for V; a piece covers
  // `subs` consecutive positions; lane l handles the piece's position
  // l/LPP, chunk l%LPP — lane-linear in LDS as glds requires. The K
  // SOURCE chunk is pre-swizzled (kswz involution) so the lane-linear
  // LDS image realizes the swizzled layout; V stays linear. Out-of-range
  // positions clamp to a valid row (their scores are masked in compute).
  auto stage_tile = [&](int t0, int b) {
    ushort_t *kimg = lds + (size_t)b * 2 * TILE_E;
    ushort_t *vimg = kimg + TILE_E;
#pragma unroll
    for (int i = 0; i < LPP / 4; ++i) {  // pieces per wave per tensor
      const int prow = (wid * (LPP / 4) + i) * subs;  // tile-local row
      const int myrow = prow + sub;
      const int pp = min(t0 + myrow, seq_len - 1);
      // IDENT: the engine's single-pool cache uses the identity page
      // table — pure address math, no table read. The general path pays
      // an ordinary VMEM load here (hipcc then drains the DMA queue at
      // its use — correct but slower; only non-identity tables take it).
      const int phys = IDENT ? (pp / page) : page_table[pp / page];
      const size_t row = ((size_t)phys * page + (pp % page)) * kh * hd +
                         (size_t)g * hd;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void
               *)(kc + row + (size_t)(sl ^ kswz(myrow)) * 8),
          (__attribute__((address_space(3))) void *)(kimg +
                                                     (size_t)prow * hd),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void *)(vc + row + sl * 8),
          (__attribute__((address_space(3))) void *)(vimg +
                                                     (size_t)prow * hd),
          16, 0, 0);
    }
  };

  // stage q (group*hd elements) by glds from wave 0 — staged, not plain
  // loads: an ordinary VMEM load whose wait lands inside the tile loop
  // forces hipcc to a loop-variant count, i.e. `vmcnt(0)` per iteration,
  // draining the DMA pipeline (.s-verified failure mode). The q pieces
  // are issued BEFORE the K/V prologue, so wave 0's first counted wait
  // covers them and the barrier publishes the image.
  typedef __attribute__((__vector_size__(8 * sizeof(short)))) short v8s;
  if (wid == 0) {
#pragma unroll
    for (int i = 0; i < (QIMG_E * 2 + 1023) / 1024; ++i) {
      if (i * 512 >= group * hd) break;  // uniform: MG may exceed group
      const int off8 = min(i * 512 + lane * 8, group * hd - 8);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void
               *)(q + (size_t)(g * group) * hd + off8),
          (__attribute__((address_space(3))) void *)(qlds + i * 512), 16, 0,
          0);
    }
  }

  const int ntiles = (limit - start + DTILE - 1) / DTILE;
  float pex[NG];
  if (ntiles > 0) {
    stage_tile(start, 0);
    if (ntiles > 1) stage_tile(start + DTILE, 1);

    for (int t = 0; t < ntiles; ++t) {
      // wait for tile t's DMA (tile t+1 stays in flight: 2 tensors x
      // LPP/4 pieces per wave = LPP/2 loads outstanding), then align.
      // Wave 0's q pieces are OLDER than tile 0's, so the same counted
      // wait covers them.
      if (ntiles > t + 1) {
        if (LPP == 16) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
        else if (LPP == 8) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
      const ushort_t *kimg = lds + (size_t)(t & 1) * 2 * TILE_E;
      const ushort_t *vimg = kimg + TILE_E;
      const int tbase = start + t * DTILE;

      // ---- QK: lane-per-position full dot (4 chunk-residue
      // accumulators cover the dependent v_dot2 latency) ----
      const bool valid = tbase + lane < limit;
      const int myswz = kswz(lane);
#pragma unroll
      for (int gi = 0; gi < NG; ++gi) {
        const int gq = wid + 4 * gi;
        if (gq >= group) break;  // wave-uniform
        float a4[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int c = 0; c < LPP; ++c) {
          const v8s kv = ((const v8s *)(kimg + (size_t)lane * hd))[c];
          const v8s qv =
              ((const v8s *)(qlds + (size_t)gq * hd))[c ^ myswz];
          a4[c & 3] = dot8_bf16(__builtin_bit_cast(bf16x8, qv),
                                __builtin_bit_cast(bf16x8, kv), a4[c & 3]);
        }
        const float s = (a4[0] + a4[1]) + (a4[2] + a4[3]);  // raw dot

        // ---- online softmax for this wave's head in the LOG2-SCALED
        // domain (m tracks m2 = max*scale*log2e; p = exp2(fma(s, scale2,
        // -m2)) — one fma+exp per element, no scale/sub passes); ONE
        // wave-wide max/sum pair per tile ----
        const float scale2 = scale * 1.4426950408889634f;
        float mx = valid ? s : -INFINITY;
        mx = wave_reduce_max(mx);
        const float mx2 = mx * scale2;
        const float mn = fmaxf(m[gi], mx2);
        const float alpha = (mn == -INFINITY) ? 0.f : exp2f(m[gi] - mn);
        const float p =
            (valid && mn != -INFINITY)
                ? exp2f(__builtin_fmaf(s, scale2, -mn))
                : 0.f;
        const float ps = wave_reduce_sum(p);
        l[gi] = l[gi] * alpha + ps;
        m[gi] = mn;
        pex[gi] = p;
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[gi][j] *= alpha;
      }

      // ---- PV: dim-sliced; this tile's P values arrive by shfl from
      // the lane that owns the position (no LDS write — see header) ----
#pragma unroll 4
      for (int it = 0; it < LPP; ++it) {
        const int p = it * subs + sub;
        const v8s vv = ((const v8s *)(vimg + (size_t)p * hd))[sl];
        const f32x8 vd = unpack8(__builtin_bit_cast(bf16x8, vv));
#pragma unroll
        for (int gi = 0; gi < NG; ++gi) {
          if (wid + 4 * gi >= group) break;
          const float pw = __shfl(pex[gi], p, WAVE);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            acc[gi][j] = fmaf(pw, vd.v[j], acc[gi][j]);
        }
      }

      // issue tile t+2 into the buffer just consumed — after re-aligning
      // so no wave still reads it
      __builtin_amdgcn_s_barrier();
      if (t + 2 < ntiles) stage_tile(tbase + 2 * DTILE, t & 1);
    }
  }

  // ---- merge the PV sub-partitions (lanes xor LPP, 2*LPP, ...) and
  // write this split's state. m/l are already wave-uniform; each wave
  // owns its head(s), so there is NO cross-wave merge at all. ----
#pragma unroll
  for (int off = LPP; off < WAVE; off <<= 1)
#pragma unroll
    for (int gi = 0; gi < NG; ++gi)
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[gi][j] += __shfl_xor(acc[gi][j], off, WAVE);

#pragma unroll
  for (int gi = 0; gi < NG; ++gi) {
    const int gq = wid + 4 * gi;
    if (gq >= group) break;
    if (n_splits == 1) {
      // whole softmax state in this block: write the output directly
      if (lane < LPP) {
        const float inv = (l[gi] > 0.f) ? 1.0f / l[gi] : 0.f;
        bf16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) o.u[j] = f32_to_bf16(acc[gi][j] * inv);
        ((bf16x8 *)(out + (size_t)(g * group + gq) * hd))[sl] = o;
      }
    } else {
      const size_t base = ((size_t)g * n_splits + split) * group + gq;
      if (lane == 0) {
        ws_m[base] = m[gi];
        ws_l[base] = l[gi];
      }
      if (lane < LPP) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          ws_acc[base * hd + sl * 8 + j] = acc[gi][j];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Decode combine kernel: grid = (hq, hd/64), block = 256 (4 waves).
// out[h, :] = sum_splits(acc * exp(m - M)) / L_total
// A SEPARATE kernel on purpose: a fused last-block-arrives combine needs
// device-scope fences, which on the 8-XCD MI355X triggered cross-L2 traffic
// that slowed the whole device ~5x (round-1 profiles).
// Parallel walk: wave w strides splits w, w+4, ...; lane = dim. The old
// 1-thread-per-(head,dim) single-block walk measured 10.5 us at 54 splits
// (62% wave-parked) — as costly as the split pass itself.
// ---------------------------------------------------------------------------

extern "C" __global__ void __launch_bounds__(256)
attn_decode_combine_kernel(const float *__restrict__ ws_m,
                           const float *__restrict__ ws_l,
                           const float *__restrict__ ws_acc,
                           ushort_t *__restrict__ out, int n_splits,
                           int group, int hd) {
  const int h = blockIdx.x;
  const int g = h / group, gi = h % group;
  const int dd = blockIdx.y * 64 + (threadIdx.x & (WAVE - 1));
  const int sg = threadIdx.x / WAVE;  // split stride group: 0..3
  if (dd >= hd) return;

  __shared__ float red[4];
  __shared__ float redL[4];

  // pass 1: global max over this wave's split stride (4 independent max
  // chains — a serial walk exposed full cross-XCD latency per entry)
  float m = -INFINITY;
  {
    float m4[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
    int s = sg;
    for (; s + 12 < n_splits; s += 16) {
#pragma unroll
      for (int u = 0; u < 4; ++u)
        m4[u] = fmaxf(m4[u],
                      ws_m[((size_t)g * n_splits + s + 4 * u) * group + gi]);
    }
    for (; s < n_splits; s += 4)
      m4[0] = fmaxf(m4[0], ws_m[((size_t)g * n_splits + s) * group + gi]);
    m = fmaxf(fmaxf(m4[0], m4[1]), fmaxf(m4[2], m4[3]));
  }
  if ((threadIdx.x & (WAVE - 1)) == 0) red[sg] = m;
  __syncthreads();
  const float M = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));

  // pass 2: strided L/acc accumulation, 4 independent accumulators per
  // wave (16 concurrent loads across the block) — the ws round-trip is
  // cross-XCD latency-bound, so ILP depth is the lever
  // ws_m holds the split kernel's LOG2-SCALED running max (m2 domain):
  // rescale factors are exp2, not exp
  float L = 0.f, Au[4] = {0.f, 0.f, 0.f, 0.f};
  int s = sg;
  for (; s + 12 < n_splits; s += 16) {
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const size_t b = ((size_t)g * n_splits + s + 4 * u) * group + gi;
      const float mw = ws_m[b];
      const float sc = (mw == -INFINITY) ? 0.f : exp2f(mw - M);
      L += ws_l[b] * sc;
      Au[u] += ws_acc[b * hd + dd] * sc;
    }
  }
  for (; s < n_splits; s += 4) {
    const size_t b = ((size_t)g * n_splits + s) * group + gi;
    const float mw = ws_m[b];
    const float sc = (mw == -INFINITY) ? 0.f : exp2f(mw - M);
    L += ws_l[b] * sc;
    Au[0] += ws_acc[b * hd + dd] * sc;
  }
  float A = (Au[0] + Au[1]) + (Au[2] + Au[3]);

  // merge the 4 waves: A via LDS columns, L via lane-0 scalars
  __shared__ float accs[4][64];
  accs[sg][threadIdx.x & (WAVE - 1)] = A;
  if ((threadIdx.x & (WAVE - 1)) == 0) redL[sg] = L;
  __syncthreads();
  if (sg == 0) {
    A = accs[0][threadIdx.x] + accs[1][threadIdx.x] + accs[2][threadIdx.x] +
        accs[3][threadIdx.x];
    L = redL[0] + redL[1] + redL[2] + redL[3];
    out[(size_t)h * hd + dd] = f32_to_bf16(L > 0.f ? A / L : 0.f);
  }
}

// ---------------------------------------------------------------------------
// Correctness-first causal prefill.
// grid.x = hq, grid.y = ceil(tq / rows_per_block), block = 256 (4 waves).
// Each 16-lane (LPP-lane) group owns ONE query row and walks keys 0..row.
// ---------------------------------------------------------------------------

template <int LPP>
__global__ void __launch_bounds__(256) attn_prefill_simple_kernel(
    const ushort_t *__restrict__ q, const ushort_t *__restrict__ k,
    const ushort_t *__restrict__ v, ushort_t *__restrict__ out,
    int tq, int tk, int kv_offset, float scale, int hq, int kh, int hd,
    int causal) {
  const int h = blockIdx.x;
  const int g = h / (hq / kh);
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int sub = lane / LPP, sl = lane % LPP;
  const int rows_per_block = 4 * (WAVE / LPP);
  const int row = blockIdx.y * rows_per_block + wid * (WAVE / LPP) + sub;
  if (row >= tq) return;

  const int limit = causal ? min(tk, kv_offset + row + 1) : tk;

  float qf[8];
  {
    const f32x8 qd =
        unpack8(((const bf16x8 *)(q + ((size_t)row * hq + h) * hd))[sl]);
#pragma unroll
    for (int j = 0; j < 8; ++j) qf[j] = qd.v[j];
  }

  float m = -INFINITY, l = 0.f, acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;

  for (int p = 0; p < limit; ++p) {
    const size_t kr = ((size_t)p * kh + g) * hd;
    const f32x8 kd = unpack8(((const bf16x8 *)(k + kr))[sl]);
    float dot = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) dot += qf[j] * kd.v[j];
    const float s = group_reduce_sum<LPP>(dot) * scale;
    const f32x8 vd = unpack8(((const bf16x8 *)(v + kr))[sl]);
    const float mn = fmaxf(m, s);
    const float alpha = __expf(m - mn);
    const float pex = __expf(s - mn);
    l = l * alpha + pex;
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = acc[j] * alpha + pex * vd.v[j];
    m = mn;
  }

  const float inv = (l > 0.f) ? 1.0f / l : 0.f;
  bf16x8 o;
#pragma unroll
  for (int j = 0; j < 8; ++j) o.u[j] = f32_to_bf16(acc[j] * inv);
  ((bf16x8 *)(out + ((size_t)row * hq + h) * hd))[sl] = o;
}

// ---------------------------------------------------------------------------
// C-linkage launch wrappers (templated LPP dispatch)
// ---------------------------------------------------------------------------

extern "C" void launch_attn_decode_split(
    const ushort_t *q, const ushort_t *kc, const ushort_t *vc,
    const int *page_table, int seq_len, float scale, int kh, int group,
    int hd, int page, int split_len, int n_splits, float *ws_m, float *ws_l,
    float *ws_acc, const int *pos_ptr, ushort_t *out, int identity,
    hipStream_t stream) {
  dim3 grid(kh, n_splits);
  const int lds = 0;  // all LDS is static in the kernel (stage + merge)
  // MG = smallest supported bound >= group keeps the per-head state arrays
  // (q fragments + online-softmax accumulators) sized to the real GQA
  // group; IDENT elides the page-table read (identity single-pool cache).
#define DISPATCH_ONE(LPP, MG)                                                  \
  do {                                                                         \
    if (identity)                                                              \
      attn_decode_split_kernel<LPP, MG, true><<<grid, 256, lds, stream>>>(     \
          q, kc, vc, page_table, seq_len, scale, kh, group, hd, page,          \
          split_len, ws_m, ws_l, ws_acc, pos_ptr, out);                        \
    else                                                                       \
      attn_decode_split_kernel<LPP, MG, false><<<grid, 256, lds, stream>>>(    \
          q, kc, vc, page_table, seq_len, scale, kh, group, hd, page,          \
          split_len, ws_m, ws_l, ws_acc, pos_ptr, out);                        \
  } while (0)
#define DISPATCH_MG(LPP)                                                       \
  do {                                                                         \
    if (group <= 2) DISPATCH_ONE(LPP, 2);                                      \
    else if (group <= 4) DISPATCH_ONE(LPP, 4);                                 \
    else DISPATCH_ONE(LPP, 8);                                                 \
  } while (0)

  switch (hd / 8) {
    case 4: DISPATCH_MG(4); break;
    case 8: DISPATCH_MG(8); break;
    case 16: DISPATCH_MG(16); break;
  }
#undef DISPATCH_MG
#undef DISPATCH_ONE
  if (n_splits > 1) {
    attn_decode_combine_kernel<<<dim3(kh * group, (hd + 63) / 64), 256, 0,
                                 stream>>>(ws_m, ws_l, ws_acc, out, n_splits,
                                           group, hd);
  }
}

extern "C" void launch_attn_prefill_simple(
    const ushort_t *q, const ushort_t *k, const ushort_t *v, ushort_t *out,
    int tq, int tk, int kv_offset, float scale, int hq, int kh, int hd,
    int causal, hipStream_t stream) {
  const int lpp = hd / 8;
  const int rows_per_block = 4 * (WAVE / lpp);
  dim3 grid(hq, (tq + rows_per_block - 1) / rows_per_block);
  switch (lpp) {
    case 4:
      attn_prefill_simple_kernel<4><<<grid, 256, 0, stream>>>(
          q, k, v, out, tq, tk, kv_offset, scale, hq, kh, hd, causal);
      break;
    case 8:
      attn_prefill_simple_kernel<8><<<grid, 256, 0, stream>>>(
          q, k, v, out, tq, tk, kv_offset, scale, hq, kh, hd, causal);
      break;
    case 16:
      attn_prefill_simple_kernel<16><<<grid, 256, 0, stream>>>(
          q, k, v, out, tq, tk, kv_offset, scale, hq, kh, hd, causal);
      break;
  }
}
