import os, sys, torch
sys.path.insert(0, "/root/repo")
from adversarial_spec_amd.engine.local import LocalEngine

def gen_ids(eng, graph):
    if graph: os.environ.pop("ADVSPEC_NO_GRAPH", None)
    else: os.environ["ADVSPEC_NO_GRAPH"] = "1"
    # wrap to capture raw ids: monkeypatch tokenizer.decode to identity-ish
    out = {}
    orig = eng.tokenizer.decode
    def cap(ids):
        out["ids"] = list(ids)
        return orig(ids)
    eng.tokenizer.decode = cap
    eng.generate("sys", "graph parity prompt", max_tokens=24, temperature=0.0, timeout=300)
    eng.tokenizer.decode = orig
    return out["ids"]

eng = LocalEngine({"name": "g3", "arch": "debug-1b"}, device="cuda:0")
a1 = gen_ids(eng, graph=False)
a2 = gen_ids(eng, graph=False)
b1 = gen_ids(eng, graph=True)
b2 = gen_ids(eng, graph=True)
print("eager1:", a1)
print("eager2:", a2)
print("graph1:", b1)
print("graph2:", b2)
for name, x, y in [("eager1-vs-eager2", a1, a2), ("eager-vs-graph", a1, b1), ("graph1-vs-graph2", b1, b2)]:
    div = next((i for i, (p, q) in enumerate(zip(x, y)) if p != q), None)
    print(name, "first divergence:", div, "lens", len(x), len(y))
