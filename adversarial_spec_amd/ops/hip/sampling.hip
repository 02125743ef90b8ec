// On-device sampling: fused temperature scale + softmax + inverse-CDF draw
// (and greedy argmax), one block over the vocab row.
//
// Decode emits one token at a time; sampling on-device avoids a logits
// round-trip (vocab 128256 x 2 B per step) — only a 4-byte token id ever
// reaches the host, and only every N tokens (engine/local.py async loop).
// The random draw is a counter-based hash of the host-provided seed, so
// replays (HIP graphs) stay deterministic.
//
// All passes read the logits as bf16x8 (16 B/lane): the row was just
// written by the lm_head GEMV so it is L2-resident; vectorized reads keep
// the single-workgroup scan latency-bound rather than issue-bound.
//
// Exact nucleus (top-p < 1) needs a sorted vocab and runs on the cold path
// in Python (ops/__init__.py); this kernel is the default temperature /
// greedy path (reference defaults 0.7 / 0.3: models.py:626).

#include "common.h"

// logits: [vocab] bf16, vocab % 8 == 0. temp <= 0 -> greedy. Returns the
// sampled id in *out; shared body for the plain and graph-state kernels.
//
// Softmax sampling via the GUMBEL-MAX trick: argmax_i(logit_i/T + g_i) with
// g_i = -log(-log(u_i)) is an exact draw from softmax(logits/T). That
// collapses the old 3-pass (max, Z, CDF-scan) sampler into ONE pass over
// the vocab — the decode step's tail latency (the lm_head logits row is
// L2-resident when this runs). Per-element RNG is a counter-based hash of
// (seed, index), so HIP-graph replays stay deterministic for a given seed.
DEVINL void sample_body(const ushort_t *__restrict__ logits, int vocab,
                        float temp, uint32_t seed, int *__restrict__ out) {
  const int tid = threadIdx.x;
  const int nvec = vocab / 8;
  const bf16x8 *lv = (const bf16x8 *)logits;
  const bool greedy = temp <= 0.f;
  const float inv_t = greedy ? 1.0f : 1.0f / temp;

  float best = -INFINITY;
  int besti = 0;
  for (int i = tid; i < nvec; i += blockDim.x) {
    const f32x8 v = unpack8(lv[i]);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int idx = i * 8 + j;
      float val = v.v[j] * inv_t;
      if (!greedy) {
        const uint32_t h = hash_u32(seed ^ ((uint32_t)idx * 2654435761u));
        const float u = ((float)h + 1.0f) * 2.3283064e-10f;  // (0, 1]
        val -= __logf(-__logf(u) + 1e-30f);  // + Gumbel(0,1)
      }
      if (val > best || (val == best && idx < besti)) { best = val; besti = idx; }
    }
  }
  // wave argmax reduce (ties -> lowest index, replay-deterministic)
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ov = __shfl_xor(best, off, WAVE);
    const int oi = __shfl_xor(besti, off, WAVE);
    if (ov > best || (ov == best && oi < besti)) { best = ov; besti = oi; }
  }
  __shared__ float mv[16];
  __shared__ int mi[16];
  const int wid = tid / WAVE;
  if ((tid & (WAVE - 1)) == 0) { mv[wid] = best; mi[wid] = besti; }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < (int)(blockDim.x / WAVE); ++w) {
      if (mv[w] > best || (mv[w] == best && mi[w] < besti)) {
        best = mv[w]; besti = mi[w];
      }
    }
    *out = besti;
  }
}

extern "C" __global__ void __launch_bounds__(1024)
sample_kernel(const ushort_t *__restrict__ logits, int vocab, float temp,
              uint32_t seed, int *__restrict__ out) {
  sample_body(logits, vocab, temp, seed, out);
}

// HIP-graph decode variant: the RNG seed, step index, TEMPERATURE and token
// outputs all live in device words so one captured graph replays per token
// and serves every request temperature (the graph cache key must not
// include temperature — ADVICE round 1).
//   temp_state: f32[1], sampling temperature (<= 0 -> greedy)
//   rng_state:  uint32[1], advanced each call
//   step_state: int32[1], the decode step index (bumped by bump_kernel)
//   tok_hist:   int32[max_new] history the host polls every N tokens
//   tok_long:   int64[1] embedding index for the next forward — written
//               directly so the decode step drops the int32->int64 copy
//               launch torch would otherwise add per token
extern "C" __global__ void __launch_bounds__(1024)
sample_state_kernel(const ushort_t *__restrict__ logits, int vocab,
                    const float *__restrict__ temp_state,
                    uint32_t *__restrict__ rng_state,
                    int *__restrict__ tok_hist,
                    const int *__restrict__ step_state,
                    long long *__restrict__ tok_long) {
  __shared__ int picked[1];
  const uint32_t seed = *rng_state;
  sample_body(logits, vocab, *temp_state, seed, picked);
  __syncthreads();
  if (threadIdx.x == 0) {
    const int tok = picked[0];
    tok_long[0] = (long long)tok;
    tok_hist[*step_state] = tok;
    *rng_state = hash_u32(seed ^ 0x6a09e667u) | 1u;  // never 0
  }
}

// pos/step bump — the single tail kernel of the captured decode step.
extern "C" __global__ void bump_kernel(int *__restrict__ pos_state,
                                       int *__restrict__ step_state) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    pos_state[0] += 1;
    step_state[0] += 1;
  }
}
