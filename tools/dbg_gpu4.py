import os, sys, torch
sys.path.insert(0, "/root/repo")
from adversarial_spec_amd.engine.local import LocalEngine

# dirty the process like the earlier tests do
pre = LocalEngine({"name": "g1", "arch": "tiny"}, device="cuda:0")
pre.generate("sys", "warmup", max_tokens=12, temperature=0.7, timeout=120)
pre2 = LocalEngine({"name": "g2", "arch": "debug-1b"}, device="cuda:0")
pre2.generate("sys", "hello", max_tokens=12, temperature=0.7, timeout=120)

eng = LocalEngine({"name": "g3", "arch": "debug-1b"}, device="cuda:0")
ids = eng.tokenizer.render_chat("sys", "graph parity prompt")
tok = torch.tensor(ids, device="cuda:0", dtype=torch.long)
with torch.cuda.stream(eng.stream):
    lg = []
    for i in range(3):
        cache = eng.model.new_cache(len(ids) + 40)
        lg.append(eng.model.prefill(tok, cache).clone())
    torch.cuda.current_stream().synchronize()
eq01 = torch.equal(lg[0], lg[1]); eq02 = torch.equal(lg[0], lg[2])
print("prefill deterministic:", eq01, eq02)
if not eq01:
    d = (lg[0].float() - lg[1].float()).abs()
    print("max diff", d.max().item(), "n diff", (d > 0).sum().item(),
          "argmaxes", lg[0].float().argmax().item(), lg[1].float().argmax().item())
# also generate-level compare in dirty process
def gen_ids(graph):
    if graph: os.environ.pop("ADVSPEC_NO_GRAPH", None)
    else: os.environ["ADVSPEC_NO_GRAPH"] = "1"
    out = {}
    orig = eng.tokenizer.decode
    eng.tokenizer.decode = lambda ids: (out.__setitem__("ids", list(ids)), orig(ids))[1]
    eng.generate("sys", "graph parity prompt", max_tokens=24, temperature=0.0, timeout=300)
    eng.tokenizer.decode = orig
    return out["ids"]
a = gen_ids(False); b = gen_ids(True)
div = next((i for i,(p,q) in enumerate(zip(a,b)) if p!=q), None)
print("eager-vs-graph divergence at", div)
print(a); print(b)
