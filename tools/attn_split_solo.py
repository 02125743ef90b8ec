"""Minimal decode-attention run for rocprofv3 (kernel-trace or PMC)."""
import sys

sys.path.insert(0, "/root/repo")
import torch  # noqa: E402

from adversarial_spec_amd import ops  # noqa: E402

torch.manual_seed(7)
kh, group, hd, page = 8, 4, 128, 256
seq = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
iters = int(sys.argv[2]) if len(sys.argv) > 2 else 200
npages = (seq + page - 1) // page + 1
kc = torch.randn(npages, page, kh, hd, device="cuda").bfloat16()
vc = torch.randn(npages, page, kh, hd, device="cuda").bfloat16()
pt = torch.arange(npages, device="cuda", dtype=torch.int32)
q = torch.randn(kh * group, hd, device="cuda").bfloat16()
pos = torch.tensor([seq - 1], dtype=torch.int32, device="cuda")
out = torch.empty(kh * group, hd, device="cuda").bfloat16()
for _ in range(iters):
    ops.attn_decode_paged(q, kc, vc, pt, seq, None, pos_state=pos, out=out)
torch.cuda.synchronize()
print("done", seq, iters)
