// Torch extension bindings for the gfx950 kernels.
//
// Contracts mirror ops/torch_ref.py exactly (shapes/dtypes documented
// there). All activations are bf16, RoPE tables f32, workspace fp32.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <cstdlib>
#include <mutex>
#include <unordered_map>
#include <vector>

using ushort_t = unsigned short;

// launch wrappers defined in the .hip translation units
extern "C" {
void launch_attn_decode_split(const ushort_t*, const ushort_t*, const ushort_t*,
                              const int*, int, float, int, int, int, int, int,
                              int, float*, float*, float*, const int*,
                              ushort_t*, int, hipStream_t);
void launch_attn_prefill_simple(const ushort_t*, const ushort_t*, const ushort_t*,
                                ushort_t*, int, int, int, float, int, int, int,
                                int, hipStream_t);
void launch_attn_prefill_mfma(const ushort_t*, const ushort_t*, const ushort_t*,
                              ushort_t*, int, int, int, float, int, int, int,
                              hipStream_t, int* ok);
void launch_attn_prefill_mfma_abl(const ushort_t*, const ushort_t*,
                                  const ushort_t*, ushort_t*, int, int, float,
                                  int, int, int, hipStream_t);
__global__ void mfma_probe_16x16x32(const ushort_t*, const ushort_t*, float*);
__global__ void rmsnorm_kernel(const ushort_t*, const ushort_t*, ushort_t*, int, float);
__global__ void add_rmsnorm_kernel(const ushort_t*, const ushort_t*, const ushort_t*,
                                   ushort_t*, ushort_t*, int, float);
__global__ void rope_kernel(ushort_t*, ushort_t*, const float*, const float*,
                            int, int, int, int, int, long, long, const int*);
__global__ void swiglu_kernel(const ushort_t*, const ushort_t*, ushort_t*, int, int, int);
__global__ void kv_write_kernel(const ushort_t*, const ushort_t*, ushort_t*, ushort_t*,
                                const int*, int, int, int, int, int, const int*);
__global__ void rope_kv_kernel(ushort_t*, ushort_t*, const ushort_t*,
                               const float*, const float*, ushort_t*, ushort_t*,
                               const int*, int, int, int, int, int, long, long,
                               long, int, const int*);
__global__ void sample_kernel(const ushort_t*, int, float, uint32_t, int*);
__global__ void sample_state_kernel(const ushort_t*, int, const float*,
                                    uint32_t*, int*, const int*, long long*);
__global__ void bump_kernel(int*, int*);
void launch_gemv(const ushort_t*, const ushort_t*, ushort_t*, int, int,
                 hipStream_t);
void launch_gemv_gateup(const ushort_t*, const ushort_t*, ushort_t*, int, int,
                        hipStream_t);
void launch_gemv_norm(const ushort_t*, const ushort_t*, const ushort_t*,
                      ushort_t*, int, int, float, hipStream_t);
void launch_gemv_res(const ushort_t*, const ushort_t*, ushort_t*, int, int,
                     hipStream_t);
void launch_gemv_gateup_norm(const ushort_t*, const ushort_t*, const ushort_t*,
                             ushort_t*, int, int, float, hipStream_t);
void launch_gemm(const ushort_t*, const ushort_t*, ushort_t*, int, int, int,
                 hipStream_t);
void launch_gemm256(const ushort_t*, const ushort_t*, ushort_t*, int, int, int,
                    hipStream_t);
void launch_gemm_fp8(const uint8_t*, const float*, const uint8_t*, const float*,
                     ushort_t*, int, int, int, hipStream_t);
void launch_gemm_fp8_256(const uint8_t*, const float*, const uint8_t*,
                         const float*, ushort_t*, int, int, int, hipStream_t);
void launch_quant_norm_fp8(const ushort_t*, const ushort_t*, uint8_t*, float*,
                           int, float, hipStream_t);
void launch_gemv_fp8_norm(const ushort_t*, const ushort_t*, const uint8_t*,
                          const float*, ushort_t*, int, int, float,
                          hipStream_t);
void launch_gemv_fp8_resl(const ushort_t*, const uint8_t*, const float*,
                          ushort_t*, int, int, hipStream_t);
void launch_gemv_fp8_gateup_norm(const ushort_t*, const ushort_t*,
                                 const uint8_t*, const float*, ushort_t*,
                                 int, int, float, hipStream_t);
void launch_gemv_fp8_res(const uint8_t*, const float*, const uint8_t*,
                         const float*, ushort_t*, int, int, hipStream_t);
void launch_gemv_fp8_gateup(const uint8_t*, const float*, const uint8_t*,
                            const float*, ushort_t*, int, int, hipStream_t);
void launch_gemv_fp8(const uint8_t*, const float*, const uint8_t*, const float*,
                     ushort_t*, int, int, hipStream_t);
void launch_quant_fp8(const ushort_t*, uint8_t*, float*, int, int, hipStream_t);
void launch_mfma_rate(const ushort_t*, float*, int, int, hipStream_t);
}

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// --------------------------------------------------------------------------
// Persistent workspace for the split-KV decode attention, keyed by the KV
// CACHE pointer: one engine owns one cache, and the engine's generate lock
// serializes all work touching it (eager kernels and graph replays alike),
// so a per-cache scratch set is race-free — and unlike stream-keyed
// scratch it cannot alias when torch's 32-entry stream pool hands two
// engines the same underlying stream. Grown buffers retire the old tensor
// into a keep-alive list because a previously captured graph may still
// replay with the old pointer.
struct StreamWS {
  torch::Tensor attn_ws;   // fp32 scratch: decode-attention m/l/acc
  std::vector<torch::Tensor> retired;
};
static std::mutex g_ws_mu;
static std::unordered_map<void*, StreamWS> g_ws;

static float* ws_f32(torch::Tensor& t, std::vector<torch::Tensor>& retired,
                     int64_t elems, const torch::Device& dev) {
  if (!t.defined() || t.numel() < elems) {
    if (t.defined()) retired.push_back(t);
    t = torch::empty({std::max<int64_t>(elems, 64)},
                     torch::TensorOptions().dtype(at::kFloat).device(dev));
  }
  return t.data_ptr<float>();
}

#define CHECK_BF16_CUDA(t)                                                 \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");                        \
  TORCH_CHECK((t).scalar_type() == at::kBFloat16, #t " must be bf16");

static const ushort_t* uptr(const torch::Tensor& t) {
  return reinterpret_cast<const ushort_t*>(t.data_ptr());
}
static ushort_t* uptr_mut(torch::Tensor& t) {
  return reinterpret_cast<ushort_t*>(t.data_ptr());
}

// --------------------------------------------------------------------------

torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor w, double eps,
                      c10::optional<torch::Tensor> out) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(w);
  auto xc = x.contiguous();
  auto wc = w.contiguous();
  const int d = xc.size(-1);
  const int t = xc.numel() / d;
  TORCH_CHECK(d % 8 == 0, "rmsnorm: d must be a multiple of 8");
  auto y = out.has_value() ? *out : torch::empty_like(xc);
  // decode-sized calls (1-4 rows): a single 256-thread block is
  // latency-bound under a loaded memory system (rocprof: 12.6 us avg at
  // 3-opponent concurrency) — 4x the threads quarters the serial depth
  const int thr = t <= 4 ? 1024 : 256;
  rmsnorm_kernel<<<t, thr, 0, cur_stream()>>>(uptr(xc), uptr(wc), uptr_mut(y),
                                              d, (float)eps);
  return y;
}

std::tuple<torch::Tensor, torch::Tensor> add_rmsnorm(
    torch::Tensor resid, torch::Tensor delta, torch::Tensor w, double eps,
    c10::optional<torch::Tensor> out_resid,
    c10::optional<torch::Tensor> out_y) {
  CHECK_BF16_CUDA(resid);
  CHECK_BF16_CUDA(delta);
  CHECK_BF16_CUDA(w);
  auto rc = resid.contiguous();
  auto dc = delta.contiguous();
  auto wc = w.contiguous();
  const int d = rc.size(-1);
  const int t = rc.numel() / d;
  TORCH_CHECK(d % 8 == 0, "add_rmsnorm: d must be a multiple of 8");
  auto r_out = out_resid.has_value() ? *out_resid : torch::empty_like(rc);
  auto y = out_y.has_value() ? *out_y : torch::empty_like(rc);
  const int thr = t <= 4 ? 1024 : 256;  // see rmsnorm
  add_rmsnorm_kernel<<<t, thr, 0, cur_stream()>>>(
      uptr(rc), uptr(dc), uptr(wc), uptr_mut(r_out), uptr_mut(y), d, (float)eps);
  return {r_out, y};
}

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor cost,
                  torch::Tensor sint, int64_t pos0) {
  CHECK_BF16_CUDA(q);
  CHECK_BF16_CUDA(k);
  // q/k may be strided VIEWS of the fused QKV output: [t, h, hd] with an
  // arbitrary row stride but contiguous head/dim dims.
  TORCH_CHECK(q.stride(2) == 1 && k.stride(2) == 1, "rope: dim contiguous");
  TORCH_CHECK(q.stride(1) == q.size(2) && k.stride(1) == k.size(2),
              "rope: head dim contiguous");
  TORCH_CHECK(cost.scalar_type() == at::kFloat, "rope: cos table f32");
  const int t = q.size(0), hq = q.size(1), hd = q.size(2);
  const int hk = k.size(1);
  TORCH_CHECK(hd % 2 == 0);
  const int waves = t * (hq + hk);
  const int blocks = (waves * 64 + 255) / 256;
  rope_kernel<<<blocks, 256, 0, cur_stream()>>>(
      uptr_mut(q), uptr_mut(k), cost.data_ptr<float>(), sint.data_ptr<float>(),
      t, hq, hk, hd, (int)pos0, q.stride(0), k.stride(0), nullptr);
}

// graph-mode RoPE: position read from a device int32 word
void rope_inplace_ds(torch::Tensor q, torch::Tensor k, torch::Tensor cost,
                     torch::Tensor sint, torch::Tensor pos_state) {
  CHECK_BF16_CUDA(q);
  const int t = q.size(0), hq = q.size(1), hd = q.size(2);
  const int hk = k.size(1);
  const int waves = t * (hq + hk);
  const int blocks = (waves * 64 + 255) / 256;
  rope_kernel<<<blocks, 256, 0, cur_stream()>>>(
      uptr_mut(q), uptr_mut(k), cost.data_ptr<float>(), sint.data_ptr<float>(),
      t, hq, hk, hd, 0, q.stride(0), k.stride(0), pos_state.data_ptr<int>());
}

// Fused RoPE + paged KV scatter: rotate q/k in place AND write the rotated
// k plus v into the page pool in one launch (saves a kernel + a K re-read
// per layer on the kernel-count-bound decode path). pos_state (int32[1] on
// device) replaces pos0 in graph mode.
void rope_kv(torch::Tensor q, torch::Tensor k, torch::Tensor v,
             torch::Tensor cost, torch::Tensor sint, torch::Tensor kc,
             torch::Tensor vc, torch::Tensor page_table, int64_t pos0,
             const c10::optional<torch::Tensor>& pos_state) {
  CHECK_BF16_CUDA(q);
  CHECK_BF16_CUDA(k);
  CHECK_BF16_CUDA(kc);
  TORCH_CHECK(q.stride(2) == 1 && k.stride(2) == 1 && v.stride(2) == 1,
              "rope_kv: dim contiguous");
  TORCH_CHECK(q.stride(1) == q.size(2) && k.stride(1) == k.size(2) &&
                  v.stride(1) == v.size(2),
              "rope_kv: head dim contiguous");
  TORCH_CHECK(cost.scalar_type() == at::kFloat, "rope_kv: cos table f32");
  TORCH_CHECK(kc.is_contiguous() && vc.is_contiguous(), "cache contiguous");
  TORCH_CHECK(page_table.scalar_type() == at::kInt);
  const int t = q.size(0), hq = q.size(1), hd = q.size(2);
  const int hk = k.size(1);
  const int page = kc.size(1);
  TORCH_CHECK(hd % 16 == 0, "rope_kv: hd % 16 == 0");
  const int waves = t * (hq + 2 * hk);
  const int* pp = pos_state.has_value() ? pos_state->data_ptr<int>() : nullptr;
  // decode (t==1): one wave per 64-thread block spreads the ~48
  // latency-bound slot-waves over 4x the CUs (the 12-block launch
  // measured 11.6 us avg under 3-opponent concurrency)
  const int thr = t <= 2 ? 64 : 256;
  const int blocks = (waves * 64 + thr - 1) / thr;
  rope_kv_kernel<<<blocks, thr, 0, cur_stream()>>>(
      uptr_mut(q), uptr_mut(k), uptr(v), cost.data_ptr<float>(),
      sint.data_ptr<float>(), uptr_mut(kc), uptr_mut(vc),
      page_table.data_ptr<int>(), t, hq, hk, hd, (int)pos0, q.stride(0),
      k.stride(0), v.stride(0), page, pp);
}

torch::Tensor swiglu(torch::Tensor gate, torch::Tensor up,
                     c10::optional<torch::Tensor> out_opt) {
  CHECK_BF16_CUDA(gate);
  CHECK_BF16_CUDA(up);
  TORCH_CHECK(gate.dim() == 2 && up.dim() == 2);
  TORCH_CHECK(gate.stride(1) == 1 && up.stride(1) == 1,
              "swiglu: last dim must be contiguous");
  TORCH_CHECK(gate.stride(0) == up.stride(0),
              "swiglu: gate/up must share a row stride (gate_up halves)");
  const int t = gate.size(0), f = gate.size(1);
  TORCH_CHECK(f % 8 == 0);
  auto out = out_opt.has_value() ? *out_opt : torch::empty({t, f}, gate.options());
  const long nvec = (long)t * f / 8;
  const int blocks = (int)std::min<long>((nvec + 255) / 256, 8192);
  swiglu_kernel<<<blocks, 256, 0, cur_stream()>>>(
      uptr(gate), uptr(up), uptr_mut(out), t, f, (int)gate.stride(0));
  return out;
}

void kv_write(torch::Tensor kc, torch::Tensor vc, torch::Tensor page_table,
              int64_t pos0, torch::Tensor k, torch::Tensor v) {
  CHECK_BF16_CUDA(kc);
  CHECK_BF16_CUDA(k);
  TORCH_CHECK(page_table.scalar_type() == at::kInt);
  auto kcc = kc;  // [np, page, kh, hd] must already be contiguous
  TORCH_CHECK(kc.is_contiguous() && vc.is_contiguous(), "cache contiguous");
  auto kq = k.contiguous();
  auto vq = v.contiguous();
  const int t = kq.size(0), kh = kq.size(1), hd = kq.size(2);
  const int page = kc.size(1);
  TORCH_CHECK((kh * hd) % 8 == 0);
  kv_write_kernel<<<t, 256, 0, cur_stream()>>>(
      uptr(kq), uptr(vq), uptr_mut(kcc), uptr_mut(vc),
      page_table.data_ptr<int>(), (int)pos0, t, kh, hd, page, nullptr);
}

void kv_write_ds(torch::Tensor kc, torch::Tensor vc, torch::Tensor page_table,
                 torch::Tensor pos_state, torch::Tensor k, torch::Tensor v) {
  auto kq = k.contiguous();
  auto vq = v.contiguous();
  const int t = kq.size(0), kh = kq.size(1), hd = kq.size(2);
  const int page = kc.size(1);
  kv_write_kernel<<<t, 256, 0, cur_stream()>>>(
      uptr(kq), uptr(vq), uptr_mut(kc), uptr_mut(vc),
      page_table.data_ptr<int>(), 0, t, kh, hd, page,
      pos_state.data_ptr<int>());
}

torch::Tensor attn_prefill(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                           double scale, bool causal, int64_t kv_offset) {
  CHECK_BF16_CUDA(q);
  CHECK_BF16_CUDA(k);
  CHECK_BF16_CUDA(v);
  auto qc = q.contiguous();
  auto kc = k.contiguous();
  auto vc = v.contiguous();
  const int tq = qc.size(0), hq = qc.size(1), hd = qc.size(2);
  const int tk = kc.size(0), kh = kc.size(1);
  TORCH_CHECK(hq % kh == 0, "GQA: hq % kh == 0");
  TORCH_CHECK(hd == 32 || hd == 64 || hd == 128, "hd in {32,64,128}");
  auto out = torch::empty_like(qc);
  // hd=128 causal takes the MFMA-tiled kernel; other head dims and the
  // non-causal path use the simple online-softmax kernel.
  int used_mfma = 0;
  if (hd == 128 && causal) {
    launch_attn_prefill_mfma(uptr(qc), uptr(kc), uptr(vc), uptr_mut(out), tq,
                             tk, (int)kv_offset, (float)scale, hq, kh, hd,
                             cur_stream(), &used_mfma);
  }
  if (!used_mfma) {
    launch_attn_prefill_simple(uptr(qc), uptr(kc), uptr(vc), uptr_mut(out), tq,
                               tk, (int)kv_offset, (float)scale, hq, kh, hd,
                               causal ? 1 : 0, cur_stream());
  }
  return out;
}

// Force the non-MFMA prefill (A/B + numerics anchor for tests).
torch::Tensor attn_prefill_simple(torch::Tensor q, torch::Tensor k,
                                  torch::Tensor v, double scale, bool causal,
                                  int64_t kv_offset) {
  CHECK_BF16_CUDA(q);
  auto qc = q.contiguous();
  auto kc = k.contiguous();
  auto vc = v.contiguous();
  const int tq = qc.size(0), hq = qc.size(1), hd = qc.size(2);
  const int tk = kc.size(0), kh = kc.size(1);
  auto out = torch::empty_like(qc);
  launch_attn_prefill_simple(uptr(qc), uptr(kc), uptr(vc), uptr_mut(out), tq,
                             tk, (int)kv_offset, (float)scale, hq, kh, hd,
                             causal ? 1 : 0, cur_stream());
  return out;
}

// prefill-attention ablation launcher (perf diagnosis only)
torch::Tensor attn_prefill_ablate(torch::Tensor q, torch::Tensor k,
                                  torch::Tensor v, double scale,
                                  int64_t abl) {
  CHECK_BF16_CUDA(q);
  auto qc = q.contiguous();
  auto kc = k.contiguous();
  auto vc = v.contiguous();
  const int tq = qc.size(0), hq = qc.size(1), hd = qc.size(2);
  const int tk = kc.size(0), kh = kc.size(1);
  TORCH_CHECK(hd == 128, "ablation probe is hd=128 only");
  auto out = torch::empty_like(qc);
  launch_attn_prefill_mfma_abl(uptr(qc), uptr(kc), uptr(vc), uptr_mut(out),
                               tq, tk, (float)scale, hq, kh, (int)abl,
                               cur_stream());
  return out;
}

// MFMA fragment-map probe: D[16,16] = A[16,32] @ B[32,16].
torch::Tensor mfma_probe16(torch::Tensor a, torch::Tensor b) {
  CHECK_BF16_CUDA(a);
  CHECK_BF16_CUDA(b);
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  TORCH_CHECK(ac.size(0) == 16 && ac.size(1) == 32);
  TORCH_CHECK(bc.size(0) == 32 && bc.size(1) == 16);
  auto d = torch::empty({16, 16}, torch::TensorOptions()
                                      .dtype(at::kFloat)
                                      .device(a.device()));
  mfma_probe_16x16x32<<<1, 64, 0, cur_stream()>>>(uptr(ac), uptr(bc),
                                                  d.data_ptr<float>());
  return d;
}

// Decode-attention split geometry. The split kernel's cost is dominated
// by per-wave SERIAL iterations (shfl-reduce + online-softmax chains), not
// KV bytes, so occupancy (total blocks) is the lever; ADVSPEC_SPLIT_BLOCKS
// overrides the target block count for A/B tuning on hardware.
static void split_geometry(long seq, int kh, int target_blocks,
                           int* n_splits, int* split_len) {
  // Target-block precedence: ADVSPEC_SPLIT_BLOCKS env (explicit A/B
  // override) > per-call target_blocks (the engine picks by its live
  // concurrency: solo decode wants ~512 blocks of self-occupancy, several
  // co-resident opponents want 256 — bench A/B under 3-way bf16
  // concurrency: 256-target 0.594 critiques/s vs 512-target 0.581; fp8
  // prefers 512) > compiled default 256.
  static const int env_blocks = [] {
    const char* e = getenv("ADVSPEC_SPLIT_BLOCKS");
    return e ? atoi(e) : 0;
  }();
  const int cap_blocks =
      env_blocks > 0 ? env_blocks : (target_blocks > 0 ? target_blocks : 256);
  const int target = std::max(1, cap_blocks / kh);
  int ns = std::max(1, std::min((int)((seq + 63) / 64), target));
  int sl = (int)((seq + ns - 1) / ns + 63) / 64 * 64;
  *n_splits = (int)((seq + sl - 1) / sl);
  *split_len = sl;
}

torch::Tensor attn_decode_paged(torch::Tensor q, torch::Tensor kc,
                                torch::Tensor vc, torch::Tensor page_table,
                                int64_t seq_len, double scale,
                                bool identity, int64_t split_blocks) {
  CHECK_BF16_CUDA(q);
  CHECK_BF16_CUDA(kc);
  auto qc = q.contiguous();
  TORCH_CHECK(kc.is_contiguous() && vc.is_contiguous());
  TORCH_CHECK(page_table.scalar_type() == at::kInt);
  const int hq = qc.size(0), hd = qc.size(1);
  const int page = kc.size(1), kh = kc.size(2);
  const int group = hq / kh;
  TORCH_CHECK(group <= 8, "GQA group <= 8");
  TORCH_CHECK(hd == 32 || hd == 64 || hd == 128);

  int n_splits, split_len;
  split_geometry(seq_len, kh, (int)split_blocks, &n_splits, &split_len);
  auto out = torch::empty({hq, hd}, qc.options());
  const long khnsg = (long)kh * n_splits * group;
  auto stream = cur_stream();
  float* ws;
  {
    std::lock_guard<std::mutex> lk(g_ws_mu);
    auto& w = g_ws[(void*)kc.data_ptr()];
    ws = ws_f32(w.attn_ws, w.retired, khnsg * (2 + hd), q.device());
  }
  launch_attn_decode_split(uptr(qc), uptr(kc), uptr(vc),
                           page_table.data_ptr<int>(), (int)seq_len,
                           (float)scale, kh, group, hd, page, split_len,
                           n_splits, ws, ws + khnsg, ws + 2 * khnsg,
                           nullptr, uptr_mut(out), identity ? 1 : 0, stream);
  return out;
}

// Async variant: write the sampled token id into out[idx] (int32, on
// device) without any host synchronization — the decode loop stays on the
// GPU and the host checks stop conditions every N tokens.
void sample_to(torch::Tensor logits, double temp, int64_t seed,
               torch::Tensor out, int64_t idx) {
  CHECK_BF16_CUDA(logits);
  TORCH_CHECK(out.scalar_type() == at::kInt && out.is_cuda());
  auto lc = logits.contiguous();
  sample_kernel<<<1, 1024, 0, cur_stream()>>>(
      uptr(lc), (int)lc.numel(), (float)temp, (uint32_t)seed,
      out.data_ptr<int>() + idx);
}

// Decode GEMV: y = x @ w for batch-1 x. Streams w once at HBM rate
// (hipBLASLt batch-1 measured 0.9-1.7 TB/s; this path targets ~5 TB/s).
torch::Tensor gemv(torch::Tensor x, torch::Tensor w,
                   c10::optional<torch::Tensor> out_opt) {
  // y = x @ w^T with w stored row-major [N, K] (HF layout)
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(w);
  TORCH_CHECK(w.dim() == 2 && w.is_contiguous());
  const int N = w.size(0), K = w.size(1);
  TORCH_CHECK((long)x.numel() == (long)K, "gemv: x numel == K");
  TORCH_CHECK(K % 8 == 0, "gemv: K % 8 == 0");
  auto xc = x.contiguous();
  auto y = out_opt.has_value()
               ? *out_opt
               : torch::empty(x.dim() == 2 ? std::vector<int64_t>{1, N}
                                           : std::vector<int64_t>{N},
                              x.options());
  launch_gemv(uptr(xc), uptr(w), uptr_mut(y), K, N, cur_stream());
  return y;
}

// Fused decode gate_up GEMV + SwiGLU: act = silu(x@Wg^T) * (x@Wu^T) with
// W = [gate | up] rows (the fused w_gate_up layout). Allocation-free.
void gemv_gateup(torch::Tensor x, torch::Tensor w, torch::Tensor act) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(w);
  CHECK_BF16_CUDA(act);
  TORCH_CHECK(w.dim() == 2 && w.is_contiguous());
  const int K = w.size(1), F2 = w.size(0);
  TORCH_CHECK(F2 % 2 == 0 && (long)x.numel() == (long)K && K % 8 == 0);
  TORCH_CHECK((long)act.numel() == (long)(F2 / 2));
  auto xc = x.contiguous();
  launch_gemv_gateup(uptr(xc), uptr(w), uptr_mut(act), K, F2 / 2,
                     cur_stream());
}

// rmsnorm fused into the GEMV prologue: y = rms(x) * ((x*wln) @ w^T).
// Replaces the separate (add_)rmsnorm launch before every consuming decode
// GEMV (see gemv.hip fusion comment).
torch::Tensor gemv_norm(torch::Tensor x, torch::Tensor wln, torch::Tensor w,
                        double eps, c10::optional<torch::Tensor> out_opt) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(wln);
  CHECK_BF16_CUDA(w);
  TORCH_CHECK(w.dim() == 2 && w.is_contiguous());
  const int N = w.size(0), K = w.size(1);
  TORCH_CHECK((long)x.numel() == (long)K, "gemv_norm: x numel == K");
  TORCH_CHECK((long)wln.numel() == (long)K, "gemv_norm: wln numel == K");
  TORCH_CHECK(K % 8 == 0, "gemv_norm: K % 8 == 0");
  TORCH_CHECK(K <= 32768, "gemv_norm: LDS xl staging caps K at 32768");
  auto xc = x.contiguous();
  auto lc = wln.contiguous();
  auto y = out_opt.has_value()
               ? *out_opt
               : torch::empty(x.dim() == 2 ? std::vector<int64_t>{1, N}
                                           : std::vector<int64_t>{N},
                              x.options());
  TORCH_CHECK((long)y.numel() == (long)N, "gemv_norm: out numel == N");
  launch_gemv_norm(uptr(xc), uptr(lc), uptr(w), uptr_mut(y), K, N,
                   (float)eps, cur_stream());
  return y;
}

// residual-add fused into the GEMV epilogue: resid += x @ w^T (in place).
// NOT valid under TP (the all-reduce needs the raw partial product).
void gemv_res(torch::Tensor x, torch::Tensor w, torch::Tensor resid) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(w);
  CHECK_BF16_CUDA(resid);
  TORCH_CHECK(w.dim() == 2 && w.is_contiguous());
  const int N = w.size(0), K = w.size(1);
  TORCH_CHECK((long)x.numel() == (long)K && K % 8 == 0);
  TORCH_CHECK((long)resid.numel() == (long)N, "gemv_res: resid numel == N");
  TORCH_CHECK(resid.is_contiguous());
  TORCH_CHECK(N <= 8192, "gemv_res: w32 kernel only (hidden <= 8192)");
  auto xc = x.contiguous();
  launch_gemv_res(uptr(xc), uptr(w), uptr_mut(resid), K, N, cur_stream());
}

// rmsnorm + gate_up GEMV + SwiGLU in one launch (decode MLP front half).
void gemv_gateup_norm(torch::Tensor x, torch::Tensor wln, torch::Tensor w,
                      double eps, torch::Tensor act) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(wln);
  CHECK_BF16_CUDA(w);
  CHECK_BF16_CUDA(act);
  TORCH_CHECK(w.dim() == 2 && w.is_contiguous());
  const int K = w.size(1), F2 = w.size(0);
  TORCH_CHECK(F2 % 2 == 0 && (long)x.numel() == (long)K && K % 8 == 0);
  TORCH_CHECK((long)wln.numel() == (long)K);
  TORCH_CHECK((long)act.numel() == (long)(F2 / 2));
  TORCH_CHECK(K <= 32768, "gemv_gateup_norm: LDS xl staging caps K at 32768");
  auto xc = x.contiguous();
  auto lc = wln.contiguous();
  launch_gemv_gateup_norm(uptr(xc), uptr(lc), uptr(w), uptr_mut(act), K,
                          F2 / 2, (float)eps, cur_stream());
}

// Tiled MFMA GEMM: C = A @ B, bf16, fp32 accumulation. Replaces library
// GEMMs on the prefill path (deterministic, workspace-free; see gemm.hip).
torch::Tensor gemm(torch::Tensor a, torch::Tensor b) {
  // C = a @ b^T with b stored row-major [N, K] (HF layout)
  CHECK_BF16_CUDA(a);
  CHECK_BF16_CUDA(b);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2, "gemm: 2-D inputs");
  TORCH_CHECK(a.size(1) == b.size(1), "gemm: K mismatch");
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  const int M = ac.size(0), K = ac.size(1), N = bc.size(0);
  auto c = torch::empty({M, N}, ac.options());
  // Deep-pipelined 256^2 8-phase kernel for prefill-sized shapes (its
  // counted-vmcnt schedule needs K % 128 == 0 and pays off when the M/N
  // tiles fill); the 128^2 kernel covers everything else.
  if (M > 128 && N >= 256 && K >= 512 && (K % 128) == 0) {
    launch_gemm256(uptr(ac), uptr(bc), uptr_mut(c), M, N, K, cur_stream());
  } else {
    launch_gemm(uptr(ac), uptr(bc), uptr_mut(c), M, N, K, cur_stream());
  }
  return c;
}

// Force a specific GEMM kernel (128 = two-barrier tile, 256 = 8-phase
// deep-pipelined) — A/B microbenches and the race screen; production code
// uses gemm()'s dispatcher.
torch::Tensor gemm_variant(torch::Tensor a, torch::Tensor b, int64_t which) {
  CHECK_BF16_CUDA(a);
  CHECK_BF16_CUDA(b);
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  const int M = ac.size(0), K = ac.size(1), N = bc.size(0);
  auto c = torch::empty({M, N}, ac.options());
  if (which == 256) {
    TORCH_CHECK(K >= 512 && K % 128 == 0, "gemm256 needs K%128==0, K>=512");
    launch_gemm256(uptr(ac), uptr(bc), uptr_mut(c), M, N, K, cur_stream());
  } else {
    launch_gemm(uptr(ac), uptr(bc), uptr_mut(c), M, N, K, cur_stream());
  }
  return c;
}

// MFMA issue-rate microbench (bf16 vs non-scaled fp8 16x16x32): the
// config-5 fp8-attention decision evidence (see mfma_rate.hip header).
void mfma_rate(torch::Tensor seed, torch::Tensor out, int64_t which,
               int64_t blocks) {
  CHECK_BF16_CUDA(seed);
  launch_mfma_rate(uptr(seed), out.data_ptr<float>(), (int)which,
                   (int)blocks, cur_stream());
}

// Row-quantize bf16 [M,K] to OCP e4m3 + per-row scale (scale = rowmax/448).
void quant_fp8(torch::Tensor x, torch::Tensor q, torch::Tensor scale) {
  CHECK_BF16_CUDA(x);
  TORCH_CHECK(q.scalar_type() == at::kByte && scale.scalar_type() == at::kFloat);
  auto xc = x.contiguous();
  const int K = xc.size(-1), M = xc.numel() / K;
  TORCH_CHECK(K % 8 == 0, "quant_fp8: K % 8 == 0");
  launch_quant_fp8(uptr(xc), q.data_ptr<uint8_t>(), scale.data_ptr<float>(),
                   M, K, cur_stream());
}

// fp8 GEMM: C = (x quantized rowwise) @ (w8 * wsc)^T; w8 row-major [N,K]
// e4m3 + per-row scale. x is quantized in here (eager prefill path).
torch::Tensor gemm_fp8(torch::Tensor x, torch::Tensor w8, torch::Tensor wsc) {
  CHECK_BF16_CUDA(x);
  TORCH_CHECK(w8.scalar_type() == at::kByte && w8.is_contiguous());
  auto xc = x.contiguous();
  const int M = xc.size(0), K = xc.size(1), N = w8.size(0);
  TORCH_CHECK(w8.size(1) == K, "gemm_fp8: K mismatch");
  auto opts8 = torch::TensorOptions().dtype(at::kByte).device(x.device());
  auto optsf = torch::TensorOptions().dtype(at::kFloat).device(x.device());
  auto x8 = torch::empty({M, (long)K}, opts8);
  auto xs = torch::empty({M}, optsf);
  launch_quant_fp8(uptr(xc), x8.data_ptr<uint8_t>(), xs.data_ptr<float>(),
                   M, K, cur_stream());
  auto c = torch::empty({M, N}, xc.options());
  // 8-phase 256^2 kernel for prefill-sized shapes (same dispatch rule as
  // the bf16 gemm); 128^2 covers edges and small shapes.
  if (M > 128 && N >= 256 && K >= 512 && (K % 128) == 0) {
    launch_gemm_fp8_256(x8.data_ptr<uint8_t>(), xs.data_ptr<float>(),
                        w8.data_ptr<uint8_t>(), wsc.data_ptr<float>(),
                        uptr_mut(c), M, N, K, cur_stream());
  } else {
    launch_gemm_fp8(x8.data_ptr<uint8_t>(), xs.data_ptr<float>(),
                    w8.data_ptr<uint8_t>(), wsc.data_ptr<float>(),
                    uptr_mut(c), M, N, K, cur_stream());
  }
  return c;
}

// fp8 decode GEMV (allocation-free: caller provides the x8/xs scratch and
// the output — all preallocated in the DecodeWorkspace for graph capture).
void gemv_fp8(torch::Tensor x, torch::Tensor w8, torch::Tensor wsc,
              torch::Tensor x8, torch::Tensor xs, torch::Tensor out) {
  CHECK_BF16_CUDA(x);
  auto xc = x.contiguous();
  const int K = xc.numel(), N = w8.size(0);
  launch_quant_fp8(uptr(xc), x8.data_ptr<uint8_t>(), xs.data_ptr<float>(),
                   1, K, cur_stream());
  launch_gemv_fp8(x8.data_ptr<uint8_t>(), xs.data_ptr<float>(),
                  w8.data_ptr<uint8_t>(), wsc.data_ptr<float>(),
                  uptr_mut(out), K, N, cur_stream());
}

// fp8 decode fusion (mirror of the bf16 gemv_norm/gemv_res fusion):

// rmsnorm + rowwise e4m3 quantize in one launch: x8 = fp8(rmsnorm(x,wln)),
// xs = the rowwise scale. Replaces (add_)rmsnorm + quant_fp8 on the fp8
// decode path.
void quant_norm_fp8(torch::Tensor x, torch::Tensor wln, torch::Tensor x8,
                    torch::Tensor xs, double eps) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(wln);
  auto xc = x.contiguous();
  const int K = xc.numel();
  TORCH_CHECK((long)wln.numel() == (long)K && K % 8 == 0);
  TORCH_CHECK(x8.numel() >= K && x8.scalar_type() == at::kByte);
  TORCH_CHECK(xs.numel() >= 1 && xs.scalar_type() == at::kFloat);
  launch_quant_norm_fp8(uptr(xc), uptr(wln), x8.data_ptr<uint8_t>(),
                        xs.data_ptr<float>(), K, (float)eps, cur_stream());
}

// pre-quantized fp8 GEMV (x8/xs from quant_norm_fp8): out = x8 @ w8^T
void gemv_fp8_q(torch::Tensor x8, torch::Tensor xs, torch::Tensor w8,
                torch::Tensor wsc, torch::Tensor out) {
  const int N = w8.size(0), K = w8.size(1);
  TORCH_CHECK(x8.numel() >= K && x8.scalar_type() == at::kByte);
  TORCH_CHECK((long)out.numel() == (long)N);
  launch_gemv_fp8(x8.data_ptr<uint8_t>(), xs.data_ptr<float>(),
                  w8.data_ptr<uint8_t>(), wsc.data_ptr<float>(),
                  uptr_mut(out), K, N, cur_stream());
}

// quantize + residual-epilogue fp8 GEMV: resid += (x @ w8^T) (in place).
// NOT valid under TP (the all-reduce needs the raw partial product).
void gemv_fp8_res(torch::Tensor x, torch::Tensor w8, torch::Tensor wsc,
                  torch::Tensor x8, torch::Tensor xs, torch::Tensor resid) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(resid);
  auto xc = x.contiguous();
  const int K = xc.numel(), N = w8.size(0);
  TORCH_CHECK((long)resid.numel() == (long)N && resid.is_contiguous());
  TORCH_CHECK(N <= 8192, "gemv_fp8_res: w32 kernel only (hidden <= 8192)");
  launch_quant_fp8(uptr(xc), x8.data_ptr<uint8_t>(), xs.data_ptr<float>(),
                   1, K, cur_stream());
  launch_gemv_fp8_res(x8.data_ptr<uint8_t>(), xs.data_ptr<float>(),
                      w8.data_ptr<uint8_t>(), wsc.data_ptr<float>(),
                      uptr_mut(resid), K, N, cur_stream());
}

// pre-quantized fused fp8 gate_up GEMV + SwiGLU
void gemv_fp8_gateup(torch::Tensor x8, torch::Tensor xs, torch::Tensor w8,
                     torch::Tensor wsc, torch::Tensor act) {
  CHECK_BF16_CUDA(act);
  const int K = w8.size(1), F2 = w8.size(0);
  TORCH_CHECK(F2 % 2 == 0 && x8.numel() >= K);
  TORCH_CHECK((long)act.numel() == (long)(F2 / 2));
  launch_gemv_fp8_gateup(x8.data_ptr<uint8_t>(), xs.data_ptr<float>(),
                         w8.data_ptr<uint8_t>(), wsc.data_ptr<float>(),
                         uptr_mut(act), K, F2 / 2, cur_stream());
}

// LDS-staged single-launch fp8 fused GEMVs (each block quantizes its own
// activation copy into LDS — removes the serial 1-block quant kernels
// from the decode critical path):

// y = rmsnorm(x, wln) @ w8^T (quantize + GEMV fused)
void gemv_fp8_norm(torch::Tensor x, torch::Tensor wln, torch::Tensor w8,
                   torch::Tensor wsc, double eps, torch::Tensor out) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(wln);
  auto xc = x.contiguous();
  const int N = w8.size(0), K = w8.size(1);
  TORCH_CHECK((long)xc.numel() == (long)K && K % 16 == 0);
  TORCH_CHECK((long)wln.numel() == (long)K);
  TORCH_CHECK((long)out.numel() == (long)N);
  TORCH_CHECK(K <= 65536, "gemv_fp8_norm: LDS staging caps K at 65536");
  launch_gemv_fp8_norm(uptr(xc), uptr(wln), w8.data_ptr<uint8_t>(),
                       wsc.data_ptr<float>(), uptr_mut(out), K, N,
                       (float)eps, cur_stream());
}

// resid += x @ w8^T (quantize + GEMV + residual epilogue, in place).
// NOT valid under TP (the all-reduce needs the raw partial product).
void gemv_fp8_resl(torch::Tensor x, torch::Tensor w8, torch::Tensor wsc,
                   torch::Tensor resid) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(resid);
  auto xc = x.contiguous();
  const int N = w8.size(0), K = w8.size(1);
  TORCH_CHECK((long)xc.numel() == (long)K && K % 16 == 0);
  TORCH_CHECK((long)resid.numel() == (long)N && resid.is_contiguous());
  TORCH_CHECK(N <= 8192, "gemv_fp8_resl: w32 kernel only (hidden <= 8192)");
  TORCH_CHECK(K <= 65536, "gemv_fp8_resl: LDS staging caps K at 65536");
  launch_gemv_fp8_resl(uptr(xc), w8.data_ptr<uint8_t>(),
                       wsc.data_ptr<float>(), uptr_mut(resid), K, N,
                       cur_stream());
}

// act = swiglu(rmsnorm(x, wln) @ [Wg|Wu]^T) — fp8 MLP front half, 1 launch
void gemv_fp8_gateup_norm(torch::Tensor x, torch::Tensor wln,
                          torch::Tensor w8, torch::Tensor wsc, double eps,
                          torch::Tensor act) {
  CHECK_BF16_CUDA(x);
  CHECK_BF16_CUDA(wln);
  CHECK_BF16_CUDA(act);
  auto xc = x.contiguous();
  const int K = w8.size(1), F2 = w8.size(0);
  TORCH_CHECK(F2 % 2 == 0 && (long)xc.numel() == (long)K && K % 16 == 0);
  TORCH_CHECK((long)wln.numel() == (long)K);
  TORCH_CHECK((long)act.numel() == (long)(F2 / 2));
  TORCH_CHECK(K <= 65536, "gemv_fp8_gateup_norm: LDS caps K at 65536");
  launch_gemv_fp8_gateup_norm(uptr(xc), uptr(wln), w8.data_ptr<uint8_t>(),
                              wsc.data_ptr<float>(), uptr_mut(act), K,
                              F2 / 2, (float)eps, cur_stream());
}

int64_t sample(torch::Tensor logits, double temp, double top_p, int64_t seed) {
  CHECK_BF16_CUDA(logits);
  TORCH_CHECK(top_p >= 1.0, "kernel sample handles top_p == 1 (nucleus is a cold path)");
  auto lc = logits.contiguous();
  const int vocab = lc.numel();
  auto out = torch::empty({1}, torch::TensorOptions()
                                   .dtype(at::kInt)
                                   .device(logits.device()));
  sample_kernel<<<1, 1024, 0, cur_stream()>>>(uptr(lc), vocab, (float)temp,
                                             (uint32_t)seed,
                                             out.data_ptr<int>());
  return out.cpu().item<int>();
}

// graph-mode decode attention: seq_len = *pos_state + 1 in-kernel; split
// count is sized once for the WHOLE generation (max_seq bound) so the
// launch geometry is replay-stable.
torch::Tensor attn_decode_paged_ds(torch::Tensor q, torch::Tensor kc,
                                   torch::Tensor vc, torch::Tensor page_table,
                                   torch::Tensor pos_state, int64_t max_seq,
                                   double scale,
                                   c10::optional<torch::Tensor> out_opt,
                                   bool identity, int64_t split_blocks) {
  CHECK_BF16_CUDA(q);
  auto qc = q.contiguous();
  const int hq = qc.size(0), hd = qc.size(1);
  const int page = kc.size(1), kh = kc.size(2);
  const int group = hq / kh;
  int n_splits, split_len;
  split_geometry(max_seq, kh, (int)split_blocks, &n_splits, &split_len);
  auto out = out_opt.has_value() ? *out_opt : torch::empty({hq, hd}, qc.options());
  const long khnsg = (long)kh * n_splits * group;
  auto stream = cur_stream();
  float* ws;
  {
    std::lock_guard<std::mutex> lk(g_ws_mu);
    auto& w = g_ws[(void*)kc.data_ptr()];
    ws = ws_f32(w.attn_ws, w.retired, khnsg * (2 + hd), q.device());
  }
  launch_attn_decode_split(uptr(qc), uptr(kc), uptr(vc),
                           page_table.data_ptr<int>(), (int)max_seq,
                           (float)scale, kh, group, hd, page, split_len,
                           n_splits, ws, ws + khnsg, ws + 2 * khnsg,
                           pos_state.data_ptr<int>(), uptr_mut(out),
                           identity ? 1 : 0, stream);
  return out;
}

void sample_state(torch::Tensor logits, torch::Tensor temp_state,
                  torch::Tensor rng_state, torch::Tensor tok_hist,
                  torch::Tensor step_state, torch::Tensor tok_long) {
  CHECK_BF16_CUDA(logits);
  TORCH_CHECK(temp_state.scalar_type() == at::kFloat && temp_state.is_cuda(),
              "sample_state: temp_state must be a device f32 word");
  TORCH_CHECK(tok_long.scalar_type() == at::kLong,
              "sample_state: tok_long must be int64 (the embedding index)");
  auto lc = logits.contiguous();
  sample_state_kernel<<<1, 1024, 0, cur_stream()>>>(
      uptr(lc), (int)lc.numel(), temp_state.data_ptr<float>(),
      reinterpret_cast<uint32_t*>(rng_state.data_ptr<int>()),
      tok_hist.data_ptr<int>(), step_state.data_ptr<int>(),
      reinterpret_cast<long long*>(tok_long.data_ptr<int64_t>()));
}

// Release the decode-attention scratch entries whose keys (per-layer KV
// cache base pointers) lie inside a dropped cache allocation, and free the
// retired grown tensors. Call ONLY after every captured graph referencing
// the cache has been destroyed (engine/local.py drops _graph_state first);
// without this, long-lived processes leak scratch per cache growth and can
// inherit a stale entry when the allocator reuses a freed cache address
// (ADVICE round 1).
void ws_release(torch::Tensor kc_all) {
  char* base = reinterpret_cast<char*>(kc_all.data_ptr());
  const size_t bytes = (size_t)kc_all.numel() * kc_all.element_size();
  std::lock_guard<std::mutex> lk(g_ws_mu);
  for (auto it = g_ws.begin(); it != g_ws.end();) {
    char* p = reinterpret_cast<char*>(it->first);
    if (p >= base && p < base + bytes) {
      it = g_ws.erase(it);
    } else {
      ++it;
    }
  }
}

void bump(torch::Tensor pos_state, torch::Tensor step_state) {
  bump_kernel<<<1, 64, 0, cur_stream()>>>(pos_state.data_ptr<int>(),
                                          step_state.data_ptr<int>());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "fused RMSNorm (bf16, gfx950)",
        py::arg("x"), py::arg("w"), py::arg("eps"),
        py::arg("out") = py::none());
  m.def("add_rmsnorm", &add_rmsnorm, "fused residual add + RMSNorm",
        py::arg("resid"), py::arg("delta"), py::arg("w"), py::arg("eps"),
        py::arg("out_resid") = py::none(), py::arg("out_y") = py::none());
  m.def("rope_inplace", &rope_inplace, "RoPE (interleaved pairs, table-driven)");
  m.def("swiglu", &swiglu, "fused silu(gate)*up",
        py::arg("gate"), py::arg("up"), py::arg("out") = py::none());
  m.def("kv_write", &kv_write, "paged KV scatter");
  m.def("attn_prefill", &attn_prefill, "causal prefill attention");
  m.def("attn_prefill_simple", &attn_prefill_simple, "non-MFMA prefill (anchor)");
  m.def("attn_prefill_ablate", &attn_prefill_ablate,
        "prefill attention perf-ablation variants");
  m.def("mfma_probe16", &mfma_probe16, "MFMA 16x16x32 fragment-map probe");
  m.def("attn_decode_paged", &attn_decode_paged, "paged decode attention",
        py::arg("q"), py::arg("kc"), py::arg("vc"), py::arg("page_table"),
        py::arg("seq_len"), py::arg("scale"), py::arg("identity") = false,
        py::arg("split_blocks") = 0);
  m.def("sample", &sample, "fused temperature softmax sample");
  m.def("sample_to", &sample_to, "async on-device sample into out[idx]");
  m.def("gemv", &gemv, "batch-1 decode GEMV (weight streaming)",
        py::arg("x"), py::arg("w"), py::arg("out") = py::none());
  m.def("gemv_gateup", &gemv_gateup, "fused gate_up GEMV + SwiGLU (decode)");
  m.def("gemv_norm", &gemv_norm, "rmsnorm-prologue GEMV (decode fusion)",
        py::arg("x"), py::arg("wln"), py::arg("w"), py::arg("eps"),
        py::arg("out") = py::none());
  m.def("gemv_res", &gemv_res, "residual-add-epilogue GEMV (decode fusion)");
  m.def("gemv_gateup_norm", &gemv_gateup_norm,
        "rmsnorm + gate_up GEMV + SwiGLU (decode fusion)");
  m.def("gemm", &gemm, "tiled MFMA GEMM (bf16, fp32 accum)");
  m.def("gemm_variant", &gemm_variant, "force GEMM kernel 128/256 (A/B)");
  m.def("gemm_fp8", &gemm_fp8, "fp8 e4m3 MFMA GEMM (rowwise scales)");
  m.def("gemv_fp8", &gemv_fp8, "fp8 decode GEMV (rowwise scales)");
  m.def("quant_norm_fp8", &quant_norm_fp8,
        "fused rmsnorm + rowwise e4m3 quantize (fp8 decode fusion)");
  m.def("gemv_fp8_q", &gemv_fp8_q, "pre-quantized fp8 decode GEMV");
  m.def("gemv_fp8_res", &gemv_fp8_res,
        "fp8 GEMV with residual-add epilogue (fp8 decode fusion)");
  m.def("gemv_fp8_gateup", &gemv_fp8_gateup,
        "pre-quantized fp8 gate_up GEMV + SwiGLU (fp8 decode fusion)");
  m.def("gemv_fp8_norm", &gemv_fp8_norm,
        "rmsnorm + quantize + fp8 GEMV, one launch (LDS-staged)");
  m.def("gemv_fp8_resl", &gemv_fp8_resl,
        "quantize + fp8 GEMV + residual epilogue, one launch (LDS-staged)");
  m.def("gemv_fp8_gateup_norm", &gemv_fp8_gateup_norm,
        "rmsnorm + quantize + fp8 gate_up GEMV + SwiGLU, one launch");
  m.def("quant_fp8", &quant_fp8, "rowwise bf16 -> e4m3 quantizer");
  m.def("mfma_rate", &mfma_rate, "MFMA issue-rate microbench (bf16/fp8)");
  m.def("rope_inplace_ds", &rope_inplace_ds, "graph-mode RoPE (device pos)");
  m.def("rope_kv", &rope_kv, "fused RoPE + paged KV scatter",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("cost"),
        py::arg("sint"), py::arg("kc"), py::arg("vc"), py::arg("page_table"),
        py::arg("pos0"), py::arg("pos_state") = py::none());
  m.def("kv_write_ds", &kv_write_ds, "graph-mode KV scatter (device pos)");
  m.def("attn_decode_paged_ds", &attn_decode_paged_ds,
        "graph-mode paged decode attention (device pos)",
        py::arg("q"), py::arg("kc"), py::arg("vc"), py::arg("page_table"),
        py::arg("pos_state"), py::arg("max_seq"), py::arg("scale"),
        py::arg("out") = py::none(), py::arg("identity") = false,
        py::arg("split_blocks") = 0);
  m.def("sample_state", &sample_state, "graph-mode on-device sampling");
  m.def("bump", &bump, "graph-mode pos/step bump");
  m.def("ws_release", &ws_release,
        "drop decode-attention scratch for a dead KV cache");
}
