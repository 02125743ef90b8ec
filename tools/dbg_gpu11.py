import os, sys, torch
sys.path.insert(0, "/root/repo")
from adversarial_spec_amd.engine.local import LocalEngine
from adversarial_spec_amd import ops
from adversarial_spec_amd.ops import _load_hip
hip = _load_hip()

g = LocalEngine({"name": "gA", "arch": "debug-1b"}, device="cuda:0")
g.generate("You are a reviewer.",
           "This is round 1 of adversarial spec development.\n\nA spec.",
           max_tokens=32, temperature=0.7, timeout=300)
g2 = LocalEngine({"name": "hA", "arch": "debug-1b"}, device="cuda:0")
g2.generate("s", "u", max_tokens=12, temperature=0.0, timeout=300)
g2.generate("s", "u", max_tokens=12, temperature=0.0, timeout=300)

eng = LocalEngine({"name": "x1", "arch": "debug-1b"}, device="cuda:0")
m = eng.model; c = m.config; dev = eng.device
rec = []
with torch.cuda.stream(eng.stream):
    ids = eng.tokenizer.render_chat("sys", "graph parity prompt")
    cache = eng._get_cache(len(ids) + 32)
    tokens = torch.tensor(ids, device=dev, dtype=torch.long)
    t = tokens.shape[0]; h, kh, hd = c.n_heads, c.n_kv_heads, c.head_dim
    resid = m.embed[tokens]
    normed = ops.rmsnorm(resid, m.layers[0].attn_norm, c.norm_eps)
    for i, L in enumerate(m.layers):
        qkv = normed @ L.wqkv
        q = qkv[:, : h * hd].view(t, h, hd)
        k = qkv[:, h * hd : (h + kh) * hd].view(t, kh, hd)
        v = qkv[:, (h + kh) * hd :].view(t, kh, hd)
        q, k = ops.rope_kv(q, k, v, m.cos, m.sin, cache.k[i], cache.v[i],
                           cache.page_table, 0)
        attn = ops.attn_prefill(q, k, v, m.scale, causal=True)
        ao = attn.reshape(t, h * hd) @ L.wo
        resid, n1 = ops.add_rmsnorm(resid, ao, L.mlp_norm, c.norm_eps)
        gu = n1 @ L.w_gate_up
        act = ops.swiglu(gu[:, : c.ffn_dim], gu[:, c.ffn_dim :])
        mo = act @ L.w_down
        nxt = m.layers[i+1].attn_norm if i+1 < c.n_layers else m.final_norm
        resid, normed = ops.add_rmsnorm(resid, mo, nxt, c.norm_eps)
        rec.append({"qkv": qkv, "q": q, "attn": attn, "ao": ao, "n1": n1,
                    "gu": gu, "act": act, "mo": mo, "resid": resid,
                    "normed": normed})
    logits = ops.gemv(normed[-1:].contiguous(), m.lm_head)
    torch.cuda.current_stream().synchronize()

print("logits nan", torch.isnan(logits.float()).sum().item())
for i, r in enumerate(rec):
    bad = {k2: torch.isnan(v2.float()).sum().item() for k2, v2 in r.items()}
    if any(bad.values()):
        print(f"L{i}:", {k2: v2 for k2, v2 in bad.items() if v2})
        if i > 3: break
