"""Decode-step cost anatomy for the bf16 8B opponent at 8k context.

Round-1 verdict item 3: solo decode ran 4.5 ms/tok vs a ~2.4-2.6 ms
weight-streaming bound (16 GB bf16 at the 6.2-6.6 TB/s the GEMV kernel
reaches). This probe accounts for the remainder per component so the fix
targets the real cost: per-kernel sums over one 32-layer step vs the
measured whole-step time (graph replay AND eager), plus the host-side
replay/launch overhead.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch  # noqa: E402

from adversarial_spec_amd.engine.local import LocalEngine  # noqa: E402


def bench(fn, iters=50, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    dtype = sys.argv[1] if len(sys.argv) > 1 else "bf16"
    spec = {"name": "probe", "arch": "llama-3-8b"}
    if dtype == "fp8":
        spec["dtype"] = "fp8"
    eng = LocalEngine(spec, device="cuda:0")
    m = eng.model
    c = m.config
    from adversarial_spec_amd.ops import _load_hip
    from adversarial_spec_amd import ops

    hip = _load_hip()

    cache = eng._get_cache(8192 + 64)
    prompt = torch.randint(0, 255, (8192,), device="cuda", dtype=torch.long)
    t0 = time.perf_counter()
    m.prefill(prompt, cache)
    torch.cuda.synchronize()
    print(f"prefill 8192: {(time.perf_counter()-t0)*1e3:.1f} ms")

    W = m.new_decode_ws()
    W.tok_long.fill_(42)
    pos = torch.tensor([8192], dtype=torch.int32, device="cuda")
    max_seq = cache.max_seq
    # mirror the engine's concurrency-aware split pick (solo here -> 512)
    m.split_blocks = eng._pick_split_blocks()
    print(f"split_blocks target: {m.split_blocks}")

    # whole step, eager
    t_step = bench(lambda: m.decode_step_ws(cache, pos, max_seq, W), iters=30)
    print(f"decode step eager: {t_step*1e3:.3f} ms/tok")

    # captured graph replay (one step per graph here: isolates replay cost)
    g = torch.cuda.CUDAGraph()
    s = torch.cuda.Stream()
    with torch.cuda.graph(g, stream=s):
        m.decode_step_ws(cache, pos, max_seq, W)
    t_graph = bench(lambda: g.replay(), iters=100)
    print(f"decode step graph replay: {t_graph*1e3:.3f} ms/tok")

    # ---- component sums over 32 layers ----
    L = m.layers[0]
    Q = m.layers_q[0] if m.fp8 else None
    h, kh, hd = c.n_heads, c.n_kv_heads, c.head_dim
    nl = c.n_layers

    def proj_t(name, out):
        if Q is not None:
            qw = Q[name]
            return bench(lambda: ops.gemv_fp8(W.normed if name != "wo" else
                                              W.attn.view(1, -1),
                                              qw.q, qw.s, W.x8, W.xs,
                                              out.view(1, -1)))
        w = getattr(L, name)
        x = W.normed if name != "wo" else W.attn.view(1, h * hd)
        return bench(lambda: ops.gemv(x, w, out=out))

    comps = {}
    comps["qkv proj"] = proj_t("wqkv", W.qkv) * nl
    comps["o proj"] = proj_t("wo", W.attn_out) * nl
    gu_w = None if Q is not None else L.w_gate_up
    if Q is not None:
        comps["gate_up proj"] = bench(lambda: ops.gemv_fp8(
            W.normed, Q["w_gate_up"].q, Q["w_gate_up"].s, W.x8, W.xs,
            W.gu.view(1, -1))) * nl
        comps["down proj"] = bench(lambda: ops.gemv_fp8(
            W.act, Q["w_down"].q, Q["w_down"].s, W.x8, W.xs,
            W.mlp_out.view(1, -1))) * nl
    else:
        comps["gate_up proj"] = bench(lambda: ops.gemv(W.normed, gu_w, out=W.gu)) * nl
        comps["down proj"] = bench(lambda: ops.gemv(W.act, L.w_down, out=W.mlp_out)) * nl

    q_ = W.qkv[:, : h * hd].view(1, h, hd)
    k_ = W.qkv[:, h * hd: (h + kh) * hd].view(1, kh, hd)
    v_ = W.qkv[:, (h + kh) * hd:].view(1, kh, hd)
    comps["rope+kv"] = bench(lambda: ops.rope_kv(
        q_, k_, v_, m.cos, m.sin, cache.k[0], cache.v[0], cache.page_table,
        0, pos_state=pos)) * nl
    comps["attention"] = bench(lambda: ops.attn_decode_paged(
        q_[0], cache.k[0], cache.v[0], cache.page_table, max_seq, m.scale,
        pos_state=pos, out=W.attn)) * nl
    comps["add_rmsnorm x2"] = bench(lambda: ops.add_rmsnorm(
        W.resid, W.attn_out, L.mlp_norm, c.norm_eps, out_resid=W.resid2,
        out_y=W.normed)) * nl * 2
    comps["swiglu"] = bench(lambda: ops.swiglu(
        W.gu[:, : c.ffn_dim], W.gu[:, c.ffn_dim:], out=W.act)) * nl
    if m.lm_head_q is not None:
        comps["lm_head"] = bench(lambda: ops.gemv_fp8(
            W.normed, m.lm_head_q.q, m.lm_head_q.s, W.x8, W.xs, W.logits))
    else:
        comps["lm_head"] = bench(lambda: ops.gemv(W.normed, m.lm_head, out=W.logits))
    comps["embed"] = bench(lambda: torch.index_select(
        m.embed, 0, W.tok_long, out=W.resid))
    tmp_t = torch.tensor([0.7], device="cuda")
    tmp_r = torch.tensor([12345], dtype=torch.int32, device="cuda")
    tmp_h = torch.zeros(64, dtype=torch.int32, device="cuda")
    tmp_s = torch.zeros(1, dtype=torch.int32, device="cuda")
    tmp_o = torch.zeros(1, dtype=torch.int64, device="cuda")
    comps["sample+bump"] = bench(lambda: hip.sample_state(
        W.logits.view(-1), tmp_t, tmp_r, tmp_h, tmp_s, tmp_o))

    total = sum(comps.values())
    print(f"\n{'component':16s} {'ms/tok':>9s}  share")
    for k, v in sorted(comps.items(), key=lambda kv: -kv[1]):
        print(f"{k:16s} {v*1e3:9.3f}  {v/total*100:5.1f}%")
    print(f"{'SUM kernels':16s} {total*1e3:9.3f}")
    print(f"{'step eager':16s} {t_step*1e3:9.3f}  (gap = launch/host = "
          f"{(t_step-total)*1e3:.3f} ms)")
    print(f"{'step graphed':16s} {t_graph*1e3:9.3f}  (gap = boundaries = "
          f"{(t_graph-total)*1e3:.3f} ms)")


if __name__ == "__main__":
    main()
