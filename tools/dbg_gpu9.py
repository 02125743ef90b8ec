import os, sys, torch
sys.path.insert(0, "/root/repo")
from adversarial_spec_amd.engine.local import LocalEngine
from adversarial_spec_amd import ops
from adversarial_spec_amd.ops import _load_hip
hip = _load_hip()

g = LocalEngine({"name": "g", "arch": "debug-1b"}, device="cuda:0")
g.generate("You are a reviewer.",
           "This is round 1 of adversarial spec development.\n\nA spec.",
           max_tokens=32, temperature=0.7, timeout=300)
g2 = LocalEngine({"name": "g2", "arch": "debug-1b"}, device="cuda:0")
g2.generate("s", "u", max_tokens=12, temperature=0.0, timeout=300)
g2.generate("s", "u", max_tokens=12, temperature=0.0, timeout=300)

def nn(x): return torch.isnan(x.float()).sum().item()

def trial(name, eng, attn_impl=None, rope_impl=None):
    m = eng.model; c = m.config; dev = eng.device
    with torch.cuda.stream(eng.stream):
        ids = eng.tokenizer.render_chat("sys", "graph parity prompt")
        cache = eng._get_cache(len(ids) + 24 + 8)
        tokens = torch.tensor(ids, device=dev, dtype=torch.long)
        t = tokens.shape[0]; h, kh, hd = c.n_heads, c.n_kv_heads, c.head_dim
        resid = m.embed[tokens]
        normed = ops.rmsnorm(resid, m.layers[0].attn_norm, c.norm_eps)
        for i, L in enumerate(m.layers):
            qkv = normed @ L.wqkv
            q = qkv[:, : h * hd].view(t, h, hd)
            k = qkv[:, h * hd : (h + kh) * hd].view(t, kh, hd)
            v = qkv[:, (h + kh) * hd :].view(t, kh, hd)
            if rope_impl == "split":
                q, k = ops.rope(q, k, m.cos, m.sin, 0)
                ops.kv_write(cache.k[i], cache.v[i], cache.page_table, 0, k, v)
            else:
                q, k = ops.rope_kv(q, k, v, m.cos, m.sin, cache.k[i], cache.v[i],
                                   cache.page_table, 0)
            if attn_impl == "simple":
                attn = hip.attn_prefill_simple(q.contiguous(), k.contiguous(),
                                               v.contiguous(), m.scale, True, 0)
            else:
                attn = ops.attn_prefill(q, k, v, m.scale, causal=True)
            ao = attn.reshape(t, h * hd) @ L.wo
            resid, normed = ops.add_rmsnorm(resid, ao, L.mlp_norm, c.norm_eps)
            gu = normed @ L.w_gate_up
            act = ops.swiglu(gu[:, : c.ffn_dim], gu[:, c.ffn_dim :])
            mo = act @ L.w_down
            nxt = m.layers[i+1].attn_norm if i+1 < c.n_layers else m.final_norm
            resid, normed = ops.add_rmsnorm(resid, mo, nxt, c.norm_eps)
        logits = ops.gemv(normed[-1:].contiguous(), m.lm_head)
        print(name, "nan:", nn(logits))

e1 = LocalEngine({"name": "x1", "arch": "debug-1b"}, device="cuda:0")
trial("async baseline     ", e1)
trial("async baseline again", e1)
trial("simple-attn        ", e1, attn_impl="simple")
trial("split rope+kv_write", e1, rope_impl="split")
trial("baseline after     ", e1)
