"""CPU numerics tests for the op reference implementations.

These are the oracle the HIP kernels are tested against (test_ops_gpu.py);
here we pin the oracle itself to hand-computed / first-principles values.
"""

import math

import pytest
import torch

from adversarial_spec_amd import ops
from adversarial_spec_amd.ops import torch_ref

torch.manual_seed(0)


class TestRMSNorm:
    def test_matches_manual(self):
        x = torch.randn(4, 64)
        w = torch.randn(64)
        y = ops.rmsnorm(x, w, 1e-5)
        for i in range(4):
            rms = math.sqrt(float((x[i] ** 2).mean()) + 1e-5)
            expected = x[i] / rms * w
            assert torch.allclose(y[i], expected, atol=1e-5)

    def test_unit_weight_unit_rows(self):
        x = torch.ones(2, 16)
        y = ops.rmsnorm(x, torch.ones(16), 0.0)
        assert torch.allclose(y, torch.ones(2, 16), atol=1e-6)


class TestAddRMSNorm:
    def test_fusion_equals_composition(self):
        resid = torch.randn(5, 32)
        delta = torch.randn(5, 32)
        w = torch.randn(32)
        r2, y = ops.add_rmsnorm(resid, delta, w, 1e-5)
        assert torch.allclose(r2, resid + delta, atol=1e-6)
        assert torch.allclose(y, ops.rmsnorm(resid + delta, w, 1e-5), atol=1e-6)


class TestRoPE:
    def test_norm_preserved(self):
        cos, sin = torch_ref.rope_tables(64, 128, 10000.0, "cpu")
        q = torch.randn(8, 4, 64)
        k = torch.randn(8, 2, 64)
        q2, k2 = ops.rope(q, k, cos, sin, 0)
        # rotation preserves the norm of each (even, odd) pair
        assert torch.allclose(q2.norm(dim=-1), q.norm(dim=-1), atol=1e-4)
        assert torch.allclose(k2.norm(dim=-1), k.norm(dim=-1), atol=1e-4)

    def test_position_zero_identity(self):
        cos, sin = torch_ref.rope_tables(32, 16, 10000.0, "cpu")
        q = torch.randn(1, 2, 32)
        k = torch.randn(1, 1, 32)
        q2, _ = ops.rope(q, k, cos, sin, 0)
        assert torch.allclose(q2, q, atol=1e-6)  # angle 0 at position 0

    def test_offset_matches_absolute(self):
        cos, sin = torch_ref.rope_tables(32, 64, 10000.0, "cpu")
        q = torch.randn(4, 2, 32)
        k = torch.randn(4, 1, 32)
        q_all, k_all = ops.rope(q, k, cos, sin, 0)
        q_off, k_off = ops.rope(q[2:], k[2:], cos, sin, 2)
        assert torch.allclose(q_off, q_all[2:], atol=1e-6)
        assert torch.allclose(k_off, k_all[2:], atol=1e-6)

    def test_relative_dot_invariance(self):
        # RoPE property: <q_m, k_n> depends only on m - n
        cos, sin = torch_ref.rope_tables(64, 128, 10000.0, "cpu")
        qv = torch.randn(1, 1, 64)
        kv = torch.randn(1, 1, 64)
        dots = []
        for m in (3, 10):
            q2, _ = ops.rope(qv, kv, cos, sin, m)
            _, k2 = ops.rope(qv, kv, cos, sin, m - 3)
            dots.append(float((q2[0, 0] * k2[0, 0]).sum()))
        assert dots[0] == pytest.approx(dots[1], abs=1e-4)


class TestSwiGLU:
    def test_formula(self):
        g = torch.randn(6, 40)
        u = torch.randn(6, 40)
        out = ops.swiglu(g, u)
        expected = torch.nn.functional.silu(g) * u
        assert torch.allclose(out, expected, atol=1e-6)


class TestAttention:
    def _naive(self, q, k, v, causal=True):
        tq, h, hd = q.shape
        tk, kh, _ = k.shape
        group = h // kh
        out = torch.zeros_like(q)
        for hh in range(h):
            kk = k[:, hh // group]
            vv = v[:, hh // group]
            for i in range(tq):
                limit = i + 1 if causal else tk
                s = (q[i, hh] @ kk[:limit].T) / math.sqrt(hd)
                p = torch.softmax(s, dim=-1)
                out[i, hh] = p @ vv[:limit]
        return out

    def test_prefill_matches_naive(self):
        q = torch.randn(7, 4, 16)
        k = torch.randn(7, 2, 16)
        v = torch.randn(7, 2, 16)
        out = ops.attn_prefill(q, k, v)
        assert torch.allclose(out, self._naive(q, k, v), atol=1e-5)

    def test_prefill_noncausal(self):
        q = torch.randn(5, 2, 8)
        k = torch.randn(5, 2, 8)
        v = torch.randn(5, 2, 8)
        out = ops.attn_prefill(q, k, v, causal=False)
        assert torch.allclose(out, self._naive(q, k, v, causal=False), atol=1e-5)

    def test_decode_matches_prefill_last_row(self):
        t = 33
        h, kh, hd = 4, 2, 16
        q = torch.randn(t, h, hd)
        k = torch.randn(t, kh, hd)
        v = torch.randn(t, kh, hd)
        full = ops.attn_prefill(q, k, v)

        # build a paged cache (page_size 8 -> crosses page boundaries)
        ps = 8
        npg = (t + ps - 1) // ps + 2
        kc = torch.zeros(npg, ps, kh, hd)
        vc = torch.zeros(npg, ps, kh, hd)
        # non-identity page table exercises the indirection
        table = torch.tensor([3, 0, 4, 2, 1, 5, 6], dtype=torch.int32)[: npg]
        ops.kv_write(kc, vc, table, 0, k, v)
        out = ops.attn_decode_paged(q[-1], kc, vc, table, t)
        assert torch.allclose(out, full[-1], atol=1e-5)

    def test_kv_write_roundtrip(self):
        ps, kh, hd = 4, 2, 8
        kc = torch.zeros(4, ps, kh, hd)
        vc = torch.zeros(4, ps, kh, hd)
        table = torch.tensor([2, 0, 3, 1], dtype=torch.int32)
        k = torch.randn(10, kh, hd)
        v = torch.randn(10, kh, hd)
        ops.kv_write(kc, vc, table, 0, k, v)
        # logical position 5 -> page 1 (physical 0), offset 1
        assert torch.allclose(kc[0, 1], k[5])
        assert torch.allclose(vc[2, 3], v[3])  # pos 3 -> page 0 (phys 2) off 3


class TestSampling:
    def test_greedy(self):
        logits = torch.tensor([0.1, 5.0, 0.2])
        assert ops.sample(logits, temperature=0.0) == 1

    def test_temperature_deterministic_with_generator(self):
        g1 = torch.Generator().manual_seed(7)
        g2 = torch.Generator().manual_seed(7)
        logits = torch.randn(100)
        a = ops.sample(logits, 0.7, generator=g1)
        b = ops.sample(logits, 0.7, generator=g2)
        assert a == b

    def test_top_p_restricts_support(self):
        logits = torch.tensor([10.0, 9.0, -50.0, -50.0])
        g = torch.Generator().manual_seed(0)
        for _ in range(20):
            assert ops.sample(logits, 1.0, top_p=0.9, generator=g) in (0, 1)

    def test_distribution_sane(self):
        logits = torch.tensor([2.0, 0.0])
        g = torch.Generator().manual_seed(3)
        counts = [0, 0]
        for _ in range(300):
            counts[ops.sample(logits, 1.0, generator=g)] += 1
        assert counts[0] > counts[1]


class TestFusedGemvCPURefs:
    """CPU reference implementations of the decode-fusion ops (the GPU
    kernels are tested in test_ops_gpu.py; these pin the CPU fallback
    contract that numerics tests compare against)."""

    EPS = 1e-5

    def _norm(self, x, wln):
        xf = x.float()
        rms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + self.EPS)
        return xf * rms * wln.float()

    def test_gemv_norm_matches_rmsnorm_then_matmul(self):
        torch.manual_seed(0)
        x = torch.randn(1, 64)
        wln = torch.rand(64) + 0.5
        w = torch.randn(32, 64)
        y = ops.gemv_norm(x, wln, w, self.EPS)
        want = self._norm(x, wln) @ w.t()
        assert torch.allclose(y.float(), want, atol=1e-5)

    def test_gemv_res_adds_in_place(self):
        torch.manual_seed(1)
        x = torch.randn(1, 64)
        w = torch.randn(32, 64)
        resid = torch.randn(1, 32)
        want = resid + x @ w.t()
        out = ops.gemv_res(x, w, resid)
        assert out is resid
        assert torch.allclose(resid, want, atol=1e-5)

    def test_gemv_gateup_norm_matches_unfused(self):
        torch.manual_seed(2)
        f = 16
        x = torch.randn(1, 64)
        wln = torch.rand(64) + 0.5
        w = torch.randn(2 * f, 64)
        out = torch.empty(1, f)
        ops.gemv_gateup_norm(x, wln, w, self.EPS, out)
        gu = self._norm(x, wln) @ w.t()
        want = torch_ref.swiglu(gu[:, :f], gu[:, f:])
        assert torch.allclose(out, want, atol=1e-5)

    def test_fp8_quant_norm_round_trip(self):
        torch.manual_seed(3)
        k = 64
        x = torch.randn(1, k)
        wln = torch.rand(k) + 0.5
        x8 = torch.empty(1, k, dtype=torch.uint8)
        xs = torch.empty(1, dtype=torch.float32)
        ops.quant_norm_fp8(x, wln, x8, xs, self.EPS)
        deq = x8.view(torch.float8_e4m3fn).float() * xs
        want = self._norm(x, wln)
        err = (deq - want).abs().max().item()
        assert err <= want.abs().max().item() * 0.08

    def test_fp8_fused_chain_cpu(self):
        torch.manual_seed(4)
        k, n, f = 64, 32, 16
        x = torch.randn(1, k)
        wln = torch.rand(k) + 0.5
        wq, wsc = ops.quantize_fp8_rowwise(torch.randn(n, k) * 0.1)
        gq, gsc = ops.quantize_fp8_rowwise(torch.randn(2 * f, k) * 0.1)

        out = torch.empty(1, n)
        ops.gemv_fp8_norm(x, wln, wq, wsc, self.EPS, out)
        assert out.abs().sum() > 0  # sane, detailed numerics on GPU

        act = torch.empty(1, f)
        ops.gemv_fp8_gateup_norm(x, wln, gq, gsc, self.EPS, act)

        resid = torch.randn(1, n)
        before = resid.clone()
        ops.gemv_fp8_resl(x, wq, wsc, resid)
        assert not torch.allclose(resid, before)

        x8 = torch.empty(1, k, dtype=torch.uint8)
        xs = torch.empty(1, dtype=torch.float32)
        ops.quant_norm_fp8(x, wln, x8, xs, self.EPS)
        out2 = torch.empty(1, n)
        ops.gemv_fp8_q(x8, xs, wq, wsc, out2)
        # pre-quantized GEMV on the quant_norm output == the one-shot op
        assert torch.allclose(out, out2, atol=1e-4)
