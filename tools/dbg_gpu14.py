import os, sys, torch
sys.path.insert(0, "/root/repo")
mode = sys.argv[1]  # graphs | nographs | freshcap
if mode == "graphs":
    os.environ["ADVSPEC_CAPTURE_ON_ENGINE"] = "1"
elif mode == "nographs":
    os.environ["ADVSPEC_NO_GRAPH"] = "1"
# freshcap: default (capture on fresh stream)
from adversarial_spec_amd.engine.local import LocalEngine

g = LocalEngine({"name": "gA", "arch": "debug-1b"}, device="cuda:0")
g.generate("You are a reviewer.",
           "This is round 1 of adversarial spec development.\n\nA spec.",
           max_tokens=32, temperature=0.7, timeout=300)
g2 = LocalEngine({"name": "hA", "arch": "debug-1b"}, device="cuda:0")
g2.generate("s", "u", max_tokens=12, temperature=0.0, timeout=300)
g2.generate("s", "u", max_tokens=12, temperature=0.0, timeout=300)

os.environ.pop("ADVSPEC_NO_GRAPH", None)
eng = LocalEngine({"name": "x1", "arch": "debug-1b"}, device="cuda:0")
m = eng.model
with torch.cuda.stream(eng.stream):
    ids = eng.tokenizer.render_chat("sys", "graph parity prompt")
    cache = eng._get_cache(len(ids) + 32)
    tok = torch.tensor(ids, device="cuda:0", dtype=torch.long)
    lg = m.prefill(tok, cache)
print(mode, "first prefill nan:", torch.isnan(lg.float()).sum().item())
