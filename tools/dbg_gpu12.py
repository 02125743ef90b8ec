import os, sys, torch
sys.path.insert(0, "/root/repo")
variant = sys.argv[1]
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
        if variant == "rope_split":
            q, k = ops.rope(q, k, m.cos, m.sin, 0)
            ops.kv_write(cache.k[i], cache.v[i], cache.page_table, 0, k, v)
        else:
            q, k = ops.rope_kv(q, k, v, m.cos, m.sin, cache.k[i], cache.v[i],
                               cache.page_table, 0)
        if variant == "attn_simple":
            attn = hip.attn_prefill_simple(q.contiguous(), k.contiguous(),
                                           v.contiguous(), m.scale, True, 0)
        elif variant == "attn_skip":
            attn = q.clone()
        else:
            attn = ops.attn_prefill(q, k, v, m.scale, causal=True)
        ao = attn.reshape(t, h * hd) @ L.wo
        if variant == "norm_ref":
            r2 = (resid.float() + ao.float())
            resid = r2.to(torch.bfloat16)
            n1 = (r2 * torch.rsqrt(r2.pow(2).mean(-1, keepdim=True) + c.norm_eps)
                  * L.mlp_norm.float()).to(torch.bfloat16)
        else:
            resid, n1 = ops.add_rmsnorm(resid, ao, L.mlp_norm, c.norm_eps)
        gu = n1 @ L.w_gate_up
        if variant == "swiglu_ref":
            gate = gu[:, : c.ffn_dim].float()
            act = (gate * torch.sigmoid(gate) * gu[:, c.ffn_dim :].float()).to(torch.bfloat16)
        else:
            act = ops.swiglu(gu[:, : c.ffn_dim], gu[:, c.ffn_dim :])
        mo = act @ L.w_down
        nxt = m.layers[i+1].attn_norm if i+1 < c.n_layers else m.final_norm
        if variant == "norm_ref":
            r2 = resid.float() + mo.float()
            resid = r2.to(torch.bfloat16)
            normed = (r2 * torch.rsqrt(r2.pow(2).mean(-1, keepdim=True) + c.norm_eps)
                      * nxt.float()).to(torch.bfloat16)
        else:
            resid, normed = ops.add_rmsnorm(resid, mo, nxt, c.norm_eps)
    logits = ops.gemv(normed[-1:].contiguous(), m.lm_head)
print(variant, "nan:", torch.isnan(logits.float()).sum().item())
