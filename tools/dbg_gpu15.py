import gc, os, sys, torch
sys.path.insert(0, "/root/repo")
mode = sys.argv[1]
from adversarial_spec_amd.engine.local import LocalEngine

def mk_g():
    g = LocalEngine({"name": "g", "arch": "debug-1b"}, device="cuda:0")
    if mode == "g_nograph":
        os.environ["ADVSPEC_NO_GRAPH"] = "1"
    g.generate("You are a reviewer.",
               "This is round 1 of adversarial spec development.\n\nA spec.",
               max_tokens=32, temperature=0.7, timeout=300)
    os.environ.pop("ADVSPEC_NO_GRAPH", None)
    return g

keep = None
if mode == "keep_alive":
    keep = mk_g()
else:
    mk_g()  # g freed here (graph + pool destroyed), like pytest scope exit
    gc.collect()

if mode == "sync_gc":
    torch.cuda.synchronize(); gc.collect(); torch.cuda.empty_cache()

g2 = LocalEngine({"name": "g2", "arch": "debug-1b"}, device="cuda:0")
with torch.cuda.stream(g2.stream):
    ids = g2.tokenizer.render_chat("s", "u")
    cache = g2._get_cache(len(ids) + 20)
    tok = torch.tensor(ids, device="cuda:0", dtype=torch.long)
    lg = g2.model.prefill(tok, cache)
n = int(torch.isnan(lg.float()).sum().item())
mx = float(lg.float().abs().max().item())
print(f"{mode}: g2 prefill nan {n} amax {mx:.3g}")
