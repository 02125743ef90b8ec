"""Decode-attention split-geometry sweep + correctness check.

The split kernel is serial-chain-bound, not KV-bandwidth-bound (round-2
anatomy: 33.7 us/layer vs ~5 us roofline at 8k). Sweeps the block-count
target (ADVSPEC_SPLIT_BLOCKS, read once per process -> child processes)
and reports us/call at decode shapes.
"""
import os
import subprocess
import sys
import time

sys.path.insert(0, "/root/repo")


def child(blocks: str) -> None:
    import torch

    from adversarial_spec_amd import ops
    from adversarial_spec_amd.ops import torch_ref

    torch.manual_seed(7)
    kh, group, hd, page = 8, 4, 128, 256
    for seq in (2048, 8192, 16384):
        npages = (seq + page - 1) // page + 1
        kc = torch.randn(npages, page, kh, hd, device="cuda").bfloat16()
        vc = torch.randn(npages, page, kh, hd, device="cuda").bfloat16()
        pt = torch.arange(npages, device="cuda", dtype=torch.int32)
        q = torch.randn(kh * group, hd, device="cuda").bfloat16()
        pos = torch.tensor([seq - 1], dtype=torch.int32, device="cuda")
        out = torch.empty(kh * group, hd, device="cuda").bfloat16()

        # correctness vs CPU fp32 reference (first call) — general path
        got = ops.attn_decode_paged(q, kc, vc, pt, seq, None, pos_state=pos,
                                    out=out)
        # identity fast path must agree bitwise (the table IS identity)
        got_id = ops.attn_decode_paged(q, kc, vc, pt, seq, None,
                                       pos_state=pos,
                                       out=torch.empty_like(out),
                                       identity=True)
        assert torch.equal(got, got_id), "identity path mismatch"
        want = torch_ref.attn_decode_paged(
            q.float().cpu(), kc.float().cpu(), vc.float().cpu(), pt.cpu(),
            seq, None)
        err = (got.float().cpu() - want).abs().max().item()
        assert err < 3e-2, f"decode attn err {err} at seq {seq}"

        def run():
            ops.attn_decode_paged(q, kc, vc, pt, seq, None, pos_state=pos,
                                  out=out, identity=True)

        for _ in range(10):
            run()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(100):
            run()
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / 100 * 1e6
        roof = 2 * seq * kh * hd * 2 / 6.4e12 * 1e6
        print(f"blocks={blocks:>5s} seq={seq:5d}: {us:7.1f} us "
              f"(roofline {roof:5.1f} us, x{us/roof:.1f})", flush=True)


if __name__ == "__main__":
    if len(sys.argv) > 1:
        child(sys.argv[1])
        sys.exit(0)
    for blocks in ("256", "512", "1024", "1536", "2048"):
        env = dict(os.environ, ADVSPEC_SPLIT_BLOCKS=blocks,
                   PYTHONPATH="/root/repo")
        r = subprocess.run([sys.executable, __file__, blocks], env=env,
                           capture_output=True, text=True, timeout=600)
        print(r.stdout, end="")
        if r.returncode != 0:
            print(r.stderr[-2000:])
