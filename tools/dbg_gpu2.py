import os, sys, torch
sys.path.insert(0, "/root/repo")
from adversarial_spec_amd.engine.local import LocalEngine
from adversarial_spec_amd import ops

eng = LocalEngine({"name": "g3", "arch": "debug-1b"}, device="cuda:0")

os.environ["ADVSPEC_NO_GRAPH"] = "1"
text, itok, otok, tm = eng.generate("sys", "graph parity prompt", max_tokens=24,
                                    temperature=0.0, timeout=300)
print("eager:", repr(text[:40]), "out_tokens", otok)
ids = eng.tokenizer.encode(text)
print("eager ids:", ids[:10])
del os.environ["ADVSPEC_NO_GRAPH"]
text2, _, otok2, _ = eng.generate("sys", "graph parity prompt", max_tokens=24,
                                  temperature=0.0, timeout=300)
print("graph:", repr(text2[:40]), "out_tokens", otok2)

# same decode by hand on the engine's stream
with torch.cuda.stream(eng.stream):
    ids0 = eng.tokenizer.render_chat("sys", "graph parity prompt")
    cache = eng.model.new_cache(len(ids0) + 40)
    tokens = torch.tensor(ids0, device="cuda:0", dtype=torch.long)
    logits = eng.model.prefill(tokens, cache)
    toks = []
    for i in range(6):
        t = int(logits.float().argmax().item())
        toks.append(t)
        logits = eng.model.decode_one(t, cache)
    print("manual stream decode:", toks, "stops:", eng.tokenizer.stop_ids())
