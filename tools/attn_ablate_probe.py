"""Prefill-attention phase ablation (guide §7 diagnostic loop, step 2).

Variants (perf-only; outputs of 1-3 are garbage by design):
  0 full | 1 -softmax | 2 -softmax -P-bounce | 3 QK^T only
Within-probe interleaved timing apportions the kernel's 250 TF/s.
"""
import sys
import time

sys.path.insert(0, "/root/repo")
import torch  # noqa: E402

from adversarial_spec_amd.ops import _load_hip  # noqa: E402

hip = _load_hip()
tq, hq, kh, hd = 8192, 32, 8, 128
q = torch.randn(tq, hq, hd, device="cuda").bfloat16()
k = torch.randn(tq, kh, hd, device="cuda").bfloat16()
v = torch.randn(tq, kh, hd, device="cuda").bfloat16()
scale = hd ** -0.5
flops = 2 * 2 * tq * tq / 2 * hd * hq


def run(abl, iters=8):
    for _ in range(2):
        hip.attn_prefill_ablate(q, k, v, scale, abl)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        hip.attn_prefill_ablate(q, k, v, scale, abl)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


names = {0: "full", 1: "-softmax", 2: "-softmax -Pstore", 3: "QK^T only"}
for _ in range(3):
    row = []
    for abl in (0, 1, 2, 3):
        t = run(abl)
        row.append(f"{names[abl]}: {t*1e3:6.2f} ms {flops/t/1e12:6.0f} TF")
    print(" | ".join(row), flush=True)
