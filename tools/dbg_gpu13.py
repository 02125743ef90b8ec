import sys, torch
sys.path.insert(0, "/root/repo")
from adversarial_spec_amd import ops
from adversarial_spec_amd.ops import _load_hip, torch_ref
hip = _load_hip()
dev = "cuda:0"

def poison():
    # fill a big chunk of the caching allocator with NaN, then free it:
    # every later torch::empty likely lands on NaN-poisoned memory.
    blocks = []
    try:
        for _ in range(40):
            blocks.append(torch.full((256, 1024, 1024), float('nan'),
                                     device=dev, dtype=torch.float32))  # 1 GiB each
    except torch.cuda.OutOfMemoryError:
        pass
    n = len(blocks)
    del blocks
    return n

def nn(x): return torch.isnan(x.float()).sum().item()

torch.manual_seed(0)
t, h, kh, hd = 56, 16, 8, 128
d = 2048; ffn = 8192
q0 = torch.randn(t, h, hd, device=dev).bfloat16()
k0 = torch.randn(t, kh, hd, device=dev).bfloat16()
v0 = torch.randn(t, kh, hd, device=dev).bfloat16()
x = torch.randn(t, d, device=dev).bfloat16()
w = torch.randn(d, ffn, device=dev).bfloat16()
wn = torch.randn(d, device=dev).bfloat16()
xv = torch.randn(1, d, device=dev).bfloat16()
wv = torch.randn(d, 128256, device=dev).bfloat16()
cos = torch.randn(4096, hd//2, device=dev); sin = torch.randn(4096, hd//2, device=dev)
kc = torch.zeros(32, 64, kh, hd, device=dev).bfloat16()
vc = torch.zeros_like(kc)
pt = torch.arange(32, dtype=torch.int32, device=dev)
torch.cuda.synchronize()
print("poisoned GiB:", poison())

for name, fn in [
    ("mfma_prefill", lambda: ops.attn_prefill(q0, k0, v0, 0.088, causal=True)),
    ("simple_prefill", lambda: hip.attn_prefill_simple(q0, k0, v0, 0.088, True, 0)),
    ("gemv", lambda: ops.gemv(xv, wv)),
    ("rmsnorm", lambda: ops.rmsnorm(x, wn, 1e-5)),
    ("add_rmsnorm", lambda: ops.add_rmsnorm(x, x, wn, 1e-5)[1]),
    ("swiglu", lambda: ops.swiglu((x @ w)[:, :ffn//2], (x @ w)[:, ffn//2:])),
    ("matmul", lambda: x @ w),
]:
    poison()
    out = fn()
    bad = nn(out)
    print(f"{name}: nan {bad}" + (f" / {out.numel()}" if bad else ""))
# rope_kv on poisoned fresh qkv-like views
poison()
qkv = (x @ torch.randn(d, (h+2*kh)*hd, device=dev).bfloat16())
q = qkv[:, : h*hd].view(t, h, hd); k = qkv[:, h*hd:(h+kh)*hd].view(t, kh, hd)
v = qkv[:, (h+kh)*hd:].view(t, kh, hd)
q, k = ops.rope_kv(q, k, v, cos, sin, kc, vc, pt, 0)
print("rope_kv: q nan", nn(q), "k", nn(k), "kc", nn(kc), "vc", nn(vc))
