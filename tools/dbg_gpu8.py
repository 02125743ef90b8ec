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
m = eng.model; c = m.config; dev = eng.device
def nn(x): return torch.isnan(x.float()).sum().item()
with torch.cuda.stream(eng.stream):
    ids = eng.tokenizer.render_chat("sys", "graph parity prompt")
    cache = eng._get_cache(len(ids) + 24 + 8)
    tokens = torch.tensor(ids, device=dev, dtype=torch.long)
    lg = m.prefill(tokens, cache)
    print("g3 prefill nan", nn(lg))
eng2 = LocalEngine({"name": "g3b", "arch": "debug-1b"}, device="cuda:0")
m = eng2.model
with torch.cuda.stream(eng2.stream):
    ids = eng2.tokenizer.render_chat("sys", "graph parity prompt")
    cache = eng2._get_cache(len(ids) + 24 + 8)
    tokens = torch.tensor(ids, device=dev, dtype=torch.long)
    lg = m.prefill(tokens, cache)
    print("g3b prefill nan", nn(lg))
    if nn(lg):
        t = tokens.shape[0]; h, kh, hd = c.n_heads, c.n_kv_heads, c.head_dim
        resid = m.embed[tokens]
        normed = ops.rmsnorm(resid, m.layers[0].attn_norm, c.norm_eps)
        L = m.layers[0]
        qkv = normed @ L.wqkv
        print("L0 qkv nan", nn(qkv))
        q = qkv[:, : h * hd].view(t, h, hd)
        k = qkv[:, h * hd : (h + kh) * hd].view(t, kh, hd)
        v = qkv[:, (h + kh) * hd :].view(t, kh, hd)
        q, k = ops.rope_kv(q, k, v, m.cos, m.sin, cache.k[0], cache.v[0],
                           cache.page_table, 0)
        print("rope q nan", nn(q), "k", nn(k), "v", nn(v))
        for trial in range(6):
            attn = ops.attn_prefill(q, k, v, m.scale, causal=True)
            a_n = nn(attn)
            print("attn trial", trial, "nan", a_n)
            if a_n:
                rows = torch.isnan(attn.float()).any(dim=2).any(dim=1).nonzero().reshape(-1)
                heads = torch.isnan(attn.float()).any(dim=2).any(dim=0).nonzero().reshape(-1)
                print("  nan rows:", rows.tolist()[:20], "heads:", heads.tolist())
        simple = ops._load_hip().attn_prefill_simple(q.contiguous(), k.contiguous(), v.contiguous(), m.scale, True, 0)
        print("simple kernel nan", nn(simple))
        ao = attn.reshape(t, h*hd) @ L.wo
        print("attn_out nan", nn(ao))
