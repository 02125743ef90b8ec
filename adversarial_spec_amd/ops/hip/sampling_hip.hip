#include "hip/hip_runtime.h"
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
DEVINL void sample_body(const ushort_t *__restrict__ logits, int vocab,
                        float temp, uint32_t seed, int *__restrict__ out) {
  __shared__ float scratch[16];
  __shared__ float tsum[256];
  __shared__ int result;
  const int tid = threadIdx.x;
  const int nvec = vocab / 8;
  const bf16x8 *lv = (const bf16x8 *)logits;

  // ---- pass 1: max (+argmax for greedy) ----
  float vmax = -INFINITY;
  int amax = 0;
  for (int i = tid; i < nvec; i += blockDim.x) {
    const f32x8 v = unpack8(lv[i]);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (v.v[j] > vmax) { vmax = v.v[j]; amax = i * 8 + j; }
    }
  }
  {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const float ov = __shfl_xor(vmax, off, WAVE);
      const int oi = __shfl_xor(amax, off, WAVE);
      if (ov > vmax || (ov == vmax && oi < amax)) { vmax = ov; amax = oi; }
    }
    __shared__ float mv[16];
    __shared__ int mi[16];
    const int wid = tid / WAVE;
    if ((tid & (WAVE - 1)) == 0) { mv[wid] = vmax; mi[wid] = amax; }
    __syncthreads();
    vmax = -INFINITY; amax = 0;
    for (int w = 0; w < (int)(blockDim.x / WAVE); ++w) {
      if (mv[w] > vmax || (mv[w] == vmax && mi[w] < amax)) {
        vmax = mv[w]; amax = mi[w];
      }
    }
  }

  if (temp <= 0.f) {
    if (tid == 0) *out = amax;
    return;
  }
  const float inv_t = 1.0f / temp;

  // ---- pass 2: Z = sum exp((v - max)/temp) ----
  float z = 0.f;
  for (int i = tid; i < nvec; i += blockDim.x) {
    const f32x8 v = unpack8(lv[i]);
#pragma unroll
    for (int j = 0; j < 8; ++j) z += __expf((v.v[j] - vmax) * inv_t);
  }
  z = block_reduce_sum(z, scratch);

  // draw u in (0, Z]; golden-ratio offset avoids the degenerate hash(0)=0
  const float target = uniform01(seed ^ 0x9e3779b9u) * z;

  // ---- pass 3: find the crossing element ----
  if (tid == 0) result = amax;  // fallback if rounding exhausts the scan
  __syncthreads();
  float running = 0.f;
  const int per_iter = blockDim.x * 8;  // elements per block iteration
  for (int base = 0; base < vocab; base += per_iter) {
    const int i = base / 8 + tid;
    float my = 0.f;
    f32x8 ev;
    if (i < nvec) {
      const f32x8 v = unpack8(lv[i]);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        ev.v[j] = __expf((v.v[j] - vmax) * inv_t);
        my += ev.v[j];
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) ev.v[j] = 0.f;
    }
    tsum[tid] = my;
    const float csum = block_reduce_sum(my, scratch);
    __syncthreads();
    if (running + csum >= target) {
      // crossing chunk: thread 0 scans the 256 per-thread sums, the owner
      // thread then pinpoints its element.
      __shared__ int owner;
      __shared__ float owner_base;
      if (tid == 0) {
        float acc = running;
        int who = blockDim.x - 1;
        for (int s = 0; s < (int)blockDim.x; ++s) {
          if (acc + tsum[s] >= target) { who = s; break; }
          acc += tsum[s];
        }
        owner = who;
        owner_base = acc;
      }
      __syncthreads();
      if (tid == owner) {
        float acc = owner_base;
        int pick = -1;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          acc += ev.v[j];
          if (acc >= target) { pick = base + tid * 8 + j; break; }
        }
        if (pick >= 0 && pick < vocab) result = pick;
      }
      break;
    }
    running += csum;
    __syncthreads();
  }
  __syncthreads();
  if (tid == 0) *out = result;
}

extern "C" __global__ void __launch_bounds__(256)
sample_kernel(const ushort_t *__restrict__ logits, int vocab, float temp,
              uint32_t seed, int *__restrict__ out) {
  sample_body(logits, vocab, temp, seed, out);
}

// HIP-graph decode variant: the RNG seed, step index and token outputs all
// live in device words so one captured graph replays per token.
//   rng_state:  uint32[1], advanced each call
//   step_state: int32[1], the decode step index (bumped by bump_kernel)
//   tok_hist:   int32[max_new] history the host polls every N tokens
//   tok_slot:   int32[1] fixed slot feeding the next embedding lookup
extern "C" __global__ void __launch_bounds__(256)
sample_state_kernel(const ushort_t *__restrict__ logits, int vocab,
                    float temp, uint32_t *__restrict__ rng_state,
                    int *__restrict__ tok_hist,
                    const int *__restrict__ step_state,
                    int *__restrict__ tok_slot) {
  __shared__ int picked[1];
  const uint32_t seed = *rng_state;
  sample_body(logits, vocab, temp, seed, picked);
  __syncthreads();
  if (threadIdx.x == 0) {
    const int tok = picked[0];
    tok_slot[0] = tok;
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
