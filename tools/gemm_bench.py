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

print("--- fp8 ---")
for M, K, N in [(8192, 4096, 6144), (8192, 4096, 28672), (56, 4096, 6144)]:
    a = torch.randn(M, K, device="cuda").bfloat16()
    w = (torch.randn(N, K, device="cuda") * 0.02).bfloat16()
    wq, wsc = ops.quantize_fp8_rowwise(w)
    t8 = bench(lambda: ops.gemm_fp8(a, wq, wsc))
    fl = 2.0 * M * K * N
    print(f"M{M} K{K} N{N}: fp8 {t8*1e3:7.2f} ms {fl/t8/1e12:7.1f} TF/s")
# decode gemv: bf16 vs fp8 (per-token weight streaming)
for K, N, tag in [(4096, 6144, "qkv"), (4096, 4096, "o"), (4096, 28672, "gate_up"), (14336, 4096, "down"), (4096, 128256, "lm")]:
    x = torch.randn(1, K, device="cuda").bfloat16()
    w = (torch.randn(N, K, device="cuda") * 0.02).bfloat16()
    wq, wsc = ops.quantize_fp8_rowwise(w)
    x8 = torch.empty(1, K, dtype=torch.uint8, device="cuda")
    xs = torch.empty(1, dtype=torch.float32, device="cuda")
    out = torch.empty(1, N, dtype=torch.bfloat16, device="cuda")
    tb = bench(lambda: ops.gemv(x, w), iters=50)
    t8 = bench(lambda: ops.gemv_fp8(x, wq, wsc, x8, xs, out), iters=50)
    by = K * N
    print(f"gemv {tag:8s}: bf16 {tb*1e6:7.1f} us {2*by/tb/1e12:5.2f} TB/s | "
          f"fp8 {t8*1e6:7.1f} us {by/t8/1e12:5.2f} TB/s")
