import torch, sys
sys.path.insert(0, "/root/repo")
from adversarial_spec_amd.engine.local import LocalEngine
from adversarial_spec_amd import ops

eng = LocalEngine({"name": "g3", "arch": "debug-1b"}, device="cuda:0")
ids = eng.tokenizer.render_chat("sys", "graph parity prompt")
print("prompt len", len(ids))
cache = eng.model.new_cache(len(ids) + 40)
tokens = torch.tensor(ids, device="cuda:0", dtype=torch.long)
logits = eng.model.prefill(tokens, cache)
lf = logits.float()
print("prefill logits: nan", torch.isnan(lf).any().item(), "inf", torch.isinf(lf).any().item(),
      "max", lf.max().item(), "argmax", lf.argmax().item())
tok = int(lf.argmax().item())
l2 = eng.model.decode_one(tok, cache).float()
print("decode logits: nan", torch.isnan(l2).any().item(), "max", l2.max().item(), "argmax", l2.argmax().item())
# sampler greedy check
s = ops.sample(logits, temperature=0.0)
print("sample greedy", s, "want", tok)
# compare one-layer decode vs torch ref? quick gemv check:
x = torch.randn(1, 2048, device="cuda:0").bfloat16()
w = torch.randn(2048, 4096, device="cuda:0").bfloat16()
y = ops.gemv(x, w).float()
want = (x.float() @ w.float())
err = (y - want).abs().max().item()
print("gemv max err", err, "rel", err / want.abs().max().item())
