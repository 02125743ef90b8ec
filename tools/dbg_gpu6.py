import os, sys, torch
sys.path.insert(0, "/root/repo")
from adversarial_spec_amd.engine.local import LocalEngine
from adversarial_spec_amd import ops

g = LocalEngine({"name": "g", "arch": "debug-1b"}, device="cuda:0")
g.generate("You are a reviewer.",
           "This is round 1 of adversarial spec development.\n\nA spec.",
           max_tokens=32, temperature=0.7, timeout=300)
g2 = LocalEngine({"name": "g2", "arch": "debug-1b"}, device="cuda:0")
g2.generate("s", "u", max_tokens=12, temperature=0.0, timeout=300)
g2.generate("s", "u", max_tokens=12, temperature=0.0, timeout=300)

eng = LocalEngine({"name": "g3", "arch": "debug-1b"}, device="cuda:0")
dev = eng.device
m = eng.model
c = m.config
with torch.cuda.stream(eng.stream):
    ids = eng.tokenizer.render_chat("sys", "graph parity prompt")
    cache = eng._get_cache(len(ids) + 24 + 8)
    tokens = torch.tensor(ids, device=dev, dtype=torch.long)
    t = tokens.shape[0]
    h, kh, hd = c.n_heads, c.n_kv_heads, c.head_dim
    resid = m.embed[tokens]
    def nan(x, tag):
        n = torch.isnan(x.float()).sum().item()
        if n: print(f"  {tag}: {n} NaNs"); return True
        return False
    nan(resid, "embed")
    normed = ops.rmsnorm(resid, m.layers[0].attn_norm, c.norm_eps)
    nan(normed, "norm0")
    for i, L in enumerate(m.layers):
        qkv = normed @ L.wqkv
        if nan(qkv, f"L{i} qkv"): break
        q = qkv[:, : h * hd].view(t, h, hd)
        k = qkv[:, h * hd : (h + kh) * hd].view(t, kh, hd)
        v = qkv[:, (h + kh) * hd :].view(t, kh, hd)
        q, k = ops.rope_kv(q, k, v, m.cos, m.sin, cache.k[i], cache.v[i],
                           cache.page_table, 0)
        if nan(q, f"L{i} q(rope)") or nan(k, f"L{i} k(rope)"): break
        attn = ops.attn_prefill(q, k, v, m.scale, causal=True)
        if nan(attn, f"L{i} attn"): break
        attn_out = attn.reshape(t, h * hd) @ L.wo
        if nan(attn_out, f"L{i} attn_out"): break
        resid, normed = ops.add_rmsnorm(resid, attn_out, L.mlp_norm, c.norm_eps)
        if nan(normed, f"L{i} post-attn norm"): break
        gu = normed @ L.w_gate_up
        if nan(gu, f"L{i} gate_up"): break
        act = ops.swiglu(gu[:, : c.ffn_dim], gu[:, c.ffn_dim :])
        if nan(act, f"L{i} swiglu"): break
        mlp_out = act @ L.w_down
        if nan(mlp_out, f"L{i} down"): break
        nxt = m.layers[i + 1].attn_norm if i + 1 < c.n_layers else m.final_norm
        resid, normed = ops.add_rmsnorm(resid, mlp_out, nxt, c.norm_eps)
        if nan(normed, f"L{i} next norm"): break
    else:
        print("no NaN found layerwise?!")
    # retry the failing op fresh if we broke out

    # lm_head gemv (the piece prefill adds beyond the layer loop)
    x = normed[-1:].contiguous()
    print("x nan", torch.isnan(x.float()).any().item())
    for trial in range(6):
        logits = ops.gemv(x, m.lm_head)
        lf = logits.float()
        n = torch.isnan(lf).sum().item()
        print(f"gemv trial {trial}: NaNs {n} argmax {lf.argmax().item()}")
        if n:
            idx = torch.isnan(lf).reshape(-1).nonzero()[:8].reshape(-1).tolist()
            print("  first nan cols:", idx)
    # torch matmul reference
    want = (x.float() @ m.lm_head.float())
    print("torch ref nan:", torch.isnan(want).any().item())
    # full prefill repeated
    for trial in range(3):
        cache2 = m.new_cache(len(ids) + 40)
        lg = m.prefill(tokens, cache2)
        print("full prefill trial", trial, "nan", torch.isnan(lg.float()).any().item())
