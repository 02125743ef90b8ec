"""Minimal fused-GEMV run for rocprofv3 --pmc (counters per dispatch).

Runs each decode-fusion kernel a few times at llama-3-8b decode shapes so
a PMC pass can attribute VALU/wait/L2 behavior per kernel without the
probe's 300-iteration timing loops.
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch  # noqa: E402

from adversarial_spec_amd import ops  # noqa: E402

DEV = "cuda:0"
EPS = 1e-5
ITERS = int(sys.argv[1]) if len(sys.argv) > 1 else 10


def bf(*shape, scale=1.0):
    return (torch.randn(*shape, device=DEV) * scale).to(torch.bfloat16)


def main():
    K, D, F, V = 4096, 4096, 14336, 128256
    x = bf(1, K)
    wln = (torch.rand(K, device=DEV) + 0.5).to(torch.bfloat16)
    resid = bf(1, D)
    w_qkv = bf(6144, K, scale=0.02)
    w_gu = bf(2 * F, K, scale=0.02)
    w_dn = bf(D, F, scale=0.02)
    w_lm = bf(V, K, scale=0.02)
    qkv = torch.empty(1, 6144, dtype=torch.bfloat16, device=DEV)
    act = torch.empty(1, F, dtype=torch.bfloat16, device=DEV)
    logits = torch.empty(1, V, dtype=torch.bfloat16, device=DEV)
    xf = bf(1, F)

    # fp8 path (quant trio default): which bound — cvt-issue or wait?
    q_qkv, s_qkv = ops.quantize_fp8_rowwise(w_qkv)
    q_gu, s_gu = ops.quantize_fp8_rowwise(w_gu)
    q_dn, s_dn = ops.quantize_fp8_rowwise(w_dn)
    x8 = torch.empty(1, max(K, F), dtype=torch.uint8, device=DEV)
    xs = torch.empty(1, dtype=torch.float32, device=DEV)

    for _ in range(ITERS):
        ops.gemv_norm(x, wln, w_qkv, EPS, out=qkv)
        ops.gemv_gateup_norm(x, wln, w_gu, EPS, act)
        ops.gemv_res(xf, w_dn, resid)
        ops.gemv_norm(x, wln, w_lm, EPS, out=logits)
        ops.quant_norm_fp8(x, wln, x8, xs, EPS)
        ops.gemv_fp8_q(x8, xs, q_qkv, s_qkv, qkv)
        ops.gemv_fp8_gateup(x8, xs, q_gu, s_gu, act)
        ops.gemv_fp8_res(xf, q_dn, s_dn, x8, xs, resid)
    torch.cuda.synchronize()
    print("pmc target done", ITERS)


if __name__ == "__main__":
    main()
