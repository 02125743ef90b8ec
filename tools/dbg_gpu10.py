import os, sys, torch
sys.path.insert(0, "/root/repo")
from adversarial_spec_amd.engine.local import LocalEngine

def pre(tag):
    g = LocalEngine({"name": "g"+tag, "arch": "debug-1b"}, device="cuda:0")
    g.generate("You are a reviewer.",
               "This is round 1 of adversarial spec development.\n\nA spec.",
               max_tokens=32, temperature=0.7, timeout=300)
    g2 = LocalEngine({"name": "h"+tag, "arch": "debug-1b"}, device="cuda:0")
    g2.generate("s", "u", max_tokens=12, temperature=0.0, timeout=300)
    g2.generate("s", "u", max_tokens=12, temperature=0.0, timeout=300)

def first_prefill_nan(tag, warm_gemm=False):
    eng = LocalEngine({"name": tag, "arch": "debug-1b"}, device="cuda:0")
    m = eng.model
    with torch.cuda.stream(eng.stream):
        if warm_gemm:
            ids0 = eng.tokenizer.render_chat("sys", "graph parity prompt")
            t = len(ids0)
            x = torch.zeros(t, m.config.dim, dtype=torch.bfloat16, device="cuda:0")
            for L in m.layers[:1]:
                (x @ L.wqkv); (x @ L.wo)
                (x @ L.w_gate_up)
                (torch.zeros(t, m.config.ffn_dim, dtype=torch.bfloat16, device="cuda:0") @ L.w_down)
            torch.cuda.current_stream().synchronize()
        ids = eng.tokenizer.render_chat("sys", "graph parity prompt")
        cache = eng._get_cache(len(ids) + 32)
        tok = torch.tensor(ids, device="cuda:0", dtype=torch.long)
        lg = m.prefill(tok, cache)
        n = torch.isnan(lg.float()).sum().item()
    print(tag, "first prefill nan:", n)

pre("A")
first_prefill_nan("x1")             # expect NaN (baseline repro)
pre("B")
first_prefill_nan("x2", warm_gemm=True)  # warmed same-shape GEMMs first
