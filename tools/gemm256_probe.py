"""Within-probe interleaved A/B: 128^2 two-barrier vs 256^2 8-phase GEMM.

Guide §5.4 rule 24: perf deltas come from interleaved rounds in ONE
process (cross-process variance > kernel deltas). Random uniform [-1,1)
operands (rule 25: zero-filled operands inflate TF via DVFS).
"""
import sys
import time

sys.path.insert(0, "/root/repo")
import torch  # noqa: E402

from adversarial_spec_amd.ops import _load_hip  # noqa: E402

hip = _load_hip()
assert hip is not None


def run(fn, iters):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


shapes = [
    (4096, 4096, 4096),
    (8192, 8192, 8192),
    (8192, 4096, 6144),    # llama-3-8b qkv
    (8192, 4096, 4096),    # o
    (8192, 4096, 28672),   # gate_up
    (8192, 14336, 4096),   # down
    (8279, 4096, 6144),    # M-edge (real prompt length)
]

for M, K, N in shapes:
    a = (torch.rand(M, K, device="cuda") * 2 - 1).bfloat16()
    b = ((torch.rand(N, K, device="cuda") * 2 - 1) * 0.05).bfloat16()
    fl = 2.0 * M * K * N
    # correctness spot-check first (fp32 GPU matmul reference)
    c256 = hip.gemm_variant(a, b, 256)
    ref = (a.float() @ b.float().t())
    err = (c256.float() - ref).abs()
    rel = (err / (ref.abs() + 1.0)).max().item()
    # interleaved A/B rounds
    t128, t256 = [], []
    iters = max(1, int(2e12 / fl))
    for _ in range(6):
        t128.append(run(lambda: hip.gemm_variant(a, b, 128), iters))
        t256.append(run(lambda: hip.gemm_variant(a, b, 256), iters))
    m128, m256 = min(t128), min(t256)
    print(f"M{M} K{K} N{N}: 128^2 {fl/m128/1e12:7.1f} TF | "
          f"256^2 {fl/m256/1e12:7.1f} TF | x{m128/m256:.3f} | relerr {rel:.2e}",
          flush=True)
    del a, b, c256, ref, err
    torch.cuda.empty_cache()
