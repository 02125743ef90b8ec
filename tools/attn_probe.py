import sys, time, torch
sys.path.insert(0, "/root/repo")
from adversarial_spec_amd.ops import _advspec_hip as hip
tq, hq, kh, hd = 8192, 32, 8, 128
q = torch.randn(tq, hq, hd, dtype=torch.bfloat16, device="cuda")
k = torch.randn(tq, kh, hd, dtype=torch.bfloat16, device="cuda")
v = torch.randn(tq, kh, hd, dtype=torch.bfloat16, device="cuda")
for _ in range(5):
    hip.attn_prefill(q, k, v, hd ** -0.5, True, 0)
torch.cuda.synchronize()
