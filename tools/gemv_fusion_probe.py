"""Per-kernel A/B: norm/res-fused decode GEMVs vs their unfused pairs.

The fusion only pays if fused_kernel <= unfused_gemv + (add_)rmsnorm for
each site; the first (mixed-stream) cut measured ~0.4 ms/step SLOWER than
unfused — this probe attributes fused-vs-unfused cost per site so a
regression is visible at the kernel level, not just in the step total.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch  # noqa: E402

from adversarial_spec_amd import ops  # noqa: E402

DEV = "cuda:0"
EPS = 1e-5


def bench(fn, iters=300, warmup=30):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def bf(*shape, scale=1.0):
    return (torch.randn(*shape, device=DEV) * scale).to(torch.bfloat16)


def main():
    K, D, F, V = 4096, 4096, 14336, 128256
    NQKV = 6144
    x = bf(1, K)
    wln = (torch.rand(K, device=DEV) + 0.5).to(torch.bfloat16)
    resid = bf(1, D)
    normed = torch.empty_like(x)
    r2 = torch.empty_like(resid)

    w_qkv = bf(NQKV, K, scale=0.02)
    w_o = bf(D, K, scale=0.02)
    w_gu = bf(2 * F, K, scale=0.02)
    w_dn = bf(D, F, scale=0.02)
    w_lm = bf(V, K, scale=0.02)
    qkv = torch.empty(1, NQKV, dtype=torch.bfloat16, device=DEV)
    act = torch.empty(1, F, dtype=torch.bfloat16, device=DEV)
    xf = bf(1, F)
    out_d = torch.empty(1, D, dtype=torch.bfloat16, device=DEV)
    logits = torch.empty(1, V, dtype=torch.bfloat16, device=DEV)

    rn = bench(lambda: ops.rmsnorm(resid, wln, EPS, out=normed))
    arn = bench(lambda: ops.add_rmsnorm(resid, out_d, wln, EPS,
                                        out_resid=r2, out_y=normed))
    print(f"rmsnorm: {rn:.2f} us   add_rmsnorm: {arn:.2f} us")

    rows = [
        ("qkv  ", bench(lambda: ops.gemv(normed, w_qkv, out=qkv)),
         bench(lambda: ops.gemv_norm(resid, wln, w_qkv, EPS, out=qkv)), rn),
        ("gateup", bench(lambda: ops.gemv_gateup(normed, w_gu, act)),
         bench(lambda: ops.gemv_gateup_norm(resid, wln, w_gu, EPS, act)), arn),
        ("o     ", bench(lambda: ops.gemv(x, w_o, out=out_d)),
         bench(lambda: ops.gemv_res(x, w_o, resid)), 0.0),
        ("down  ", bench(lambda: ops.gemv(xf, w_dn, out=out_d)),
         bench(lambda: ops.gemv_res(xf, w_dn, resid)), arn),
        ("lmhead", bench(lambda: ops.gemv(normed, w_lm, out=logits)),
         bench(lambda: ops.gemv_norm(resid, wln, w_lm, EPS, out=logits)), rn),
    ]
    print(f"{'site':7s} {'unfused':>9s} {'fused':>9s} {'norm-kern':>10s} "
          f"{'delta/site':>11s}")
    tot = 0.0
    for name, un, fu, nk in rows:
        d = fu - (un + nk)
        tot += d
        print(f"{name:7s} {un:8.2f}u {fu:8.2f}u {nk:9.2f}u {d:+10.2f}u")
    print(f"net per-layer-ish delta (neg = fusion wins): {tot:+.2f} us")


if __name__ == "__main__":
    main()
