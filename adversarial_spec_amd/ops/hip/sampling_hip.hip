#include "hip/hip_runtime.h"
// On-device sampling: fused temperature scale + softmax + inverse-CDF draw
// (and greedy argmax), one block over the vocab row.
//
// Decode emits one token at a time; doing the sample on-device avoids a
// logits round-trip to the host (vocab 128256 x 2 B each step) — only the
// 4-byte token id crosses PCIe. The random draw is a counter-based hash of
// the host-provided seed, so replays (HIP graphs) stay deterministic.
//
// Exact nucleus (top-p < 1) requires a sorted vocab and runs on the cold
// path in Python (ops/__init__.py); the kernel handles the default
// temperature/greedy paths (reference defaults: temp 0.7 / 0.3,
// BASELINE.md sampling row).

#include "common.h"

// logits: [vocab] bf16. out: int32[1]. temp <= 0 -> greedy.
extern "C" __global__ void __launch_bounds__(256)
sample_kernel(const ushort_t *__restrict__ logits, int vocab, float temp,
              uint32_t seed, int *__restrict__ out) {
  __shared__ float scratch[16];
  __shared__ float chunk_vals[256];
  __shared__ int result;
  const int tid = threadIdx.x;

  // pass 1: max (argmax for greedy)
  float vmax = -INFINITY;
  int amax = 0;
  for (int i = tid; i < vocab; i += blockDim.x) {
    const float v = bf16_to_f32(logits[i]);
    if (v > vmax) { vmax = v; amax = i; }
  }
  // block argmax via LDS pairs
  __shared__ float mv[16];
  __shared__ int mi[16];
  {
    // wave-level argmax
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const float ov = __shfl_xor(vmax, off, WAVE);
      const int oi = __shfl_xor(amax, off, WAVE);
      if (ov > vmax || (ov == vmax && oi < amax)) { vmax = ov; amax = oi; }
    }
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

  // pass 2: sum of exp((v - vmax)/temp)
  float z = 0.f;
  for (int i = tid; i < vocab; i += blockDim.x) {
    z += __expf((bf16_to_f32(logits[i]) - vmax) * inv_t);
  }
  z = block_reduce_sum(z, scratch);

  // draw u in (0, Z]; golden-ratio offset avoids the degenerate hash(0)=0
  const float target = uniform01(seed ^ 0x9e3779b9u) * z;

  // pass 3: find the crossing chunk, then scan inside it
  if (tid == 0) result = amax;  // fallback: rounding may exhaust the loop
  __syncthreads();
  float running = 0.f;
  for (int base = 0; base < vocab; base += blockDim.x) {
    const int i = base + tid;
    const float e =
        (i < vocab) ? __expf((bf16_to_f32(logits[i]) - vmax) * inv_t) : 0.f;
    chunk_vals[tid] = e;
    float csum = block_reduce_sum(e, scratch);
    __syncthreads();
    if (running + csum >= target) {
      if (tid == 0) {
        float acc = running;
        int pick = -1;
        for (int j = 0; j < (int)blockDim.x && base + j < vocab; ++j) {
          acc += chunk_vals[j];
          if (acc >= target) { pick = base + j; break; }
        }
        result = (pick >= 0) ? pick : amax;
      }
      break;
    }
    running += csum;
    __syncthreads();
  }
  __syncthreads();
  if (tid == 0) *out = result;
}
