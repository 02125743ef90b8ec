import os, sys, torch
sys.path.insert(0, "/root/repo")
from adversarial_spec_amd.engine.local import LocalEngine

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
def wnan():
    bad = []
    for nm, t in [("embed", m.embed), ("lm_head", m.lm_head), ("final", m.final_norm)]:
        if torch.isnan(t.float()).any().item(): bad.append(nm)
    for i, L in enumerate(m.layers):
        for f in ("attn_norm", "wqkv", "wo", "mlp_norm", "w_gate_up", "w_down"):
            t = getattr(L, f)
            n = torch.isnan(t.float()).sum().item()
            if n: bad.append(f"L{i}.{f}:{n}")
    return bad
print("weights nan BEFORE:", wnan())
with torch.cuda.stream(eng.stream):
    ids = eng.tokenizer.render_chat("sys", "graph parity prompt")
    cache = eng._get_cache(len(ids) + 24 + 8)
    tokens = torch.tensor(ids, device=dev, dtype=torch.long)
    for trial in range(4):
        cache.seq_len = 0
        logits = m.prefill(tokens, cache)
        print("prefill trial", trial, "nan", torch.isnan(logits.float()).sum().item())
print("weights nan AFTER:", wnan())
# now bisect within forward on the NaN-producing FIRST call... rerun fresh engine
eng2 = LocalEngine({"name": "g3b", "arch": "debug-1b"}, device="cuda:0")
with torch.cuda.stream(eng2.stream):
    cache = eng2._get_cache(len(ids) + 24 + 8)
    t2 = torch.tensor(eng2.tokenizer.render_chat("sys", "graph parity prompt"),
                      device=dev, dtype=torch.long)
    lg = eng2.model.prefill(t2, cache)
    print("g3b prefill nan", torch.isnan(lg.float()).sum().item())
    # check intermediate: which layer's cache content went NaN
    for i in range(eng2.model.config.n_layers):
        kn = torch.isnan(cache.k[i].float()).sum().item()
        vn = torch.isnan(cache.v[i].float()).sum().item()
        if kn or vn: print(f"  cache L{i}: k {kn} v {vn} NaNs")
