"""Microbench: in-tree MFMA GEMM vs torch (rocBLAS) on model shapes."""
import sys, time
sys.path.insert(0, "/root/repo")
import torch
from adversarial_spec_amd import ops

def bench(fn, iters=20):
    for _ in range(3): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

shapes = [  # (M, K, N) llama-3-8b prefill projections
    (8192, 4096, 6144),   # qkv
    (8192, 4096, 4096),   # o
    (8192, 4096, 28672),  # gate_up
    (8192, 14336, 4096),  # down
    (56, 4096, 6144),     # skinny prompt
]
for M, K, N in shapes:
    a = torch.randn(M, K, device="cuda").bfloat16()
    b = (torch.randn(N, K, device="cuda") * 0.02).bfloat16()
    t_ours = bench(lambda: ops.gemm(a, b))
    t_lib = bench(lambda: a @ b.t())
    fl = 2.0 * M * K * N
    print(f"M{M} K{K} N{N}: ours {t_ours*1e3:7.2f} ms {fl/t_ours/1e12:7.1f} TF/s"
          f" | lib {t_lib*1e3:7.2f} ms {fl/t_lib/1e12:7.1f} TF/s")
