#!/usr/bin/env python3
"""Focused GPU perf probes: prefill attention TF/s, decode tok/s by mode,
GEMV TB/s. Run on MI355X:  gpurun -- 'python tools/perf_probe.py'."""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch  # noqa: E402


def bench_gpu(fn, iters=20, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def probe_prefill_attn():
    from adversarial_spec_amd.ops import _advspec_hip as hip

    tq, hq, kh, hd = 8192, 32, 8, 128
    q = torch.randn(tq, hq, hd, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(tq, kh, hd, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(tq, kh, hd, dtype=torch.bfloat16, device="cuda")
    scale = hd ** -0.5
    flops = 2 * 2 * tq * tq / 2 * hd * hq  # causal QK^T + PV

    t = bench_gpu(lambda: hip.attn_prefill(q, k, v, scale, True, 0))
    print(f"prefill mfma   tq={tq}: {t*1e3:8.2f} ms  {flops/t/1e12:7.1f} TF/s")
    t = bench_gpu(lambda: hip.attn_prefill_simple(q, k, v, scale, True, 0), iters=3)
    print(f"prefill simple tq={tq}: {t*1e3:8.2f} ms  {flops/t/1e12:7.1f} TF/s")


def probe_gemv():
    from adversarial_spec_amd.ops import _advspec_hip as hip

    for K, N, tag in [(4096, 6144, "qkv"), (4096, 4096, "o"),
                      (4096, 28672, "gate_up"), (14336, 4096, "down"),
                      (4096, 128256, "lm_head")]:
        x = torch.randn(1, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
        t = bench_gpu(lambda: hip.gemv(x, w, None), iters=50)
        tb = K * N * 2 / t / 1e12
        tt = bench_gpu(lambda: x @ w.t(), iters=50)
        print(f"gemv {tag:8s} [{K}x{N}]: {t*1e6:7.1f} us {tb:5.2f} TB/s "
              f"(hipBLASLt {tt*1e6:7.1f} us {K*N*2/tt/1e12:5.2f} TB/s)")


def probe_decode():
    from adversarial_spec_amd.engine.local import LocalEngine

    eng = LocalEngine({"name": "probe", "arch": "llama-3-8b"}, device="cuda:0")
    sysm = "You are a reviewer."
    user = "This is round 1.\n\n" + ("spec text " * 800)

    for mode, env in [("graph", None), ("eager", "1")]:
        if env:
            os.environ["ADVSPEC_NO_GRAPH"] = env
        else:
            os.environ.pop("ADVSPEC_NO_GRAPH", None)
        # warm
        eng.generate(sysm, user, max_tokens=8, temperature=0.7, timeout=600)
        t0 = time.perf_counter()
        text, itok, otok, tm = eng.generate(
            sysm, user, max_tokens=128, temperature=0.7, timeout=600
        )
        wall = time.perf_counter() - t0
        print(f"decode[{mode}]: prefill {tm['prefill']:.0f} ms ({itok} tok), "
              f"decode {tm['decode']:.0f} ms ({otok} tok, "
              f"{tm['decode']/max(otok,1):.2f} ms/tok), wall {wall:.2f} s")




def probe_decode_attn():
    from adversarial_spec_amd.ops import _advspec_hip as hip

    kh, hd, group = 8, 128, 4
    page = 256
    for seq in (8192, 16384):
        npg = (seq + page - 1) // page + 1
        kc = torch.randn(npg, page, kh, hd, dtype=torch.bfloat16, device="cuda")
        vc = torch.randn_like(kc)
        pt = torch.arange(npg, dtype=torch.int32, device="cuda")
        q = torch.randn(kh * group, hd, dtype=torch.bfloat16, device="cuda")
        t = bench_gpu(lambda: hip.attn_decode_paged(q, kc, vc, pt, seq, 0.088),
                      iters=100)
        by = seq * kh * hd * 2 * 2
        print(f"decode-attn seq={seq}: {t*1e6:7.1f} us {by/t/1e12:5.2f} TB/s")

if __name__ == "__main__":
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    probe_gemv()
    probe_prefill_attn()
    probe_decode_attn()
    probe_decode()
