"""GPU numerics tests: every HIP kernel vs the fp32 torch reference.

Run on MI355X via:  gpurun -- 'python -m pytest tests -m gpu -x -q'
Inputs are bf16 (the kernels' wire dtype); the reference computes in fp32
on the same bf16-rounded values, so tolerances cover bf16 output rounding
plus fp32-accumulation-order differences only.
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from adversarial_spec_amd import ops
    from adversarial_spec_amd.ops import torch_ref
else:  # collected but deselected on CPU
    ops = torch_ref = None

DEV = "cuda:0"


def _bf(x):
    return x.to(torch.bfloat16)


def _assert_close(got, want, atol, rtol=2e-2, name=""):
    g = got.float().cpu()
    w = want.float().cpu()
    err = (g - w).abs()
    bound = atol + rtol * w.abs()
    frac_bad = (err > bound).float().mean().item()
    assert frac_bad == 0.0, (
        f"{name}: max err {err.max():.4g} vs bound; {frac_bad*100:.2f}% out of tol"
    )


@pytest.fixture(autouse=True)
def _seed():
    torch.manual_seed(1234)


class TestMFMAProbe:
    def test_fragment_map(self):
        """Pins the 16x16x32 A/B/C fragment maps vs torch.matmul."""
        from adversarial_spec_amd.ops import _advspec_hip

        a = _bf(torch.randn(16, 32)).to(DEV)
        b = _bf(torch.randn(32, 16)).to(DEV)
        d = _advspec_hip.mfma_probe16(a, b)
        want = a.float() @ b.float()
        _assert_close(d, want, atol=1e-2, name="mfma_probe")

    def test_fragment_map_asymmetric(self):
        """Asymmetric structured B catches transposed outputs (guide G9)."""
        from adversarial_spec_amd.ops import _advspec_hip

        a = torch.zeros(16, 32)
        a[3, 5] = 1.0
        b = torch.zeros(32, 16)
        b[5, 11] = 2.0
        d = _advspec_hip.mfma_probe16(_bf(a).to(DEV), _bf(b).to(DEV)).cpu()
        assert d[3, 11].item() == pytest.approx(2.0, abs=1e-2)
        assert d.abs().sum().item() == pytest.approx(2.0, abs=1e-2)


class TestNorms:
    def test_rmsnorm(self):
        x = _bf(torch.randn(33, 4096)).to(DEV)
        w = _bf(torch.randn(4096)).to(DEV)
        got = ops.rmsnorm(x, w, 1e-5)
        want = torch_ref.rmsnorm(x.cpu(), w.cpu(), 1e-5)
        _assert_close(got, want, atol=2e-2, name="rmsnorm")

    def test_add_rmsnorm(self):
        r = _bf(torch.randn(17, 2048)).to(DEV)
        d = _bf(torch.randn(17, 2048)).to(DEV)
        w = _bf(torch.rand(2048) + 0.5).to(DEV)
        r2, y = ops.add_rmsnorm(r, d, w, 1e-5)
        r2_ref, y_ref = torch_ref.add_rmsnorm(r.cpu(), d.cpu(), w.cpu(), 1e-5)
        _assert_close(r2, r2_ref, atol=2e-2, name="add_rmsnorm resid")
        _assert_close(y, y_ref, atol=2e-2, name="add_rmsnorm y")

    def test_rmsnorm_row_independence(self):
        x = _bf(torch.randn(4, 256)).to(DEV)
        w = _bf(torch.ones(256)).to(DEV)
        full = ops.rmsnorm(x, w, 1e-5)
        row = ops.rmsnorm(x[2:3].contiguous(), w, 1e-5)
        assert torch.equal(full[2:3], row)


class TestRoPE:
    @pytest.mark.parametrize("hd", [32, 64, 128])
    def test_vs_ref(self, hd):
        t, hq, hk = 9, 4, 2
        cos, sin = torch_ref.rope_tables(hd, 64, 10000.0, DEV)
        q = _bf(torch.randn(t, hq, hd)).to(DEV)
        k = _bf(torch.randn(t, hk, hd)).to(DEV)
        q_ref, k_ref = torch_ref.rope(q.cpu(), k.cpu(), cos.cpu(), sin.cpu(), 3)
        q2, k2 = ops.rope(q, k, cos, sin, 3)
        _assert_close(q2, q_ref, atol=2e-2, name=f"rope q hd={hd}")
        _assert_close(k2, k_ref, atol=2e-2, name=f"rope k hd={hd}")


class TestSwiGLU:
    def test_fused_halves(self):
        t, f = 13, 1024
        gu = _bf(torch.randn(t, 2 * f)).to(DEV)
        got = ops.swiglu(gu[:, :f], gu[:, f:])
        want = torch_ref.swiglu(gu[:, :f].cpu(), gu[:, f:].cpu())
        _assert_close(got, want, atol=2e-2, name="swiglu")


class TestKVWrite:
    def test_paged_scatter(self):
        kh, hd, ps, npg = 2, 128, 16, 8
        kc = torch.zeros(npg, ps, kh, hd, dtype=torch.bfloat16, device=DEV)
        vc = torch.zeros_like(kc)
        table = torch.tensor([5, 2, 7, 0, 1, 3, 4, 6], dtype=torch.int32, device=DEV)
        k = _bf(torch.randn(40, kh, hd)).to(DEV)
        v = _bf(torch.randn(40, kh, hd)).to(DEV)
        ops.kv_write(kc, vc, table, 3, k, v)
        kc_ref = torch.zeros(npg, ps, kh, hd)
        vc_ref = torch.zeros(npg, ps, kh, hd)
        torch_ref.kv_write(kc_ref, vc_ref, table.cpu(), 3, k.float().cpu(), v.float().cpu())
        assert torch.equal(kc.float().cpu(), kc_ref)
        assert torch.equal(vc.float().cpu(), vc_ref)


class TestDecodeAttention:
    @pytest.mark.parametrize("seq_len", [1, 7, 255, 256, 1000])
    @pytest.mark.parametrize("hd,group", [(128, 4), (64, 8), (32, 2)])
    def test_vs_ref(self, seq_len, hd, group):
        kh = 2
        hq = kh * group
        ps = 64
        npg = (seq_len + ps - 1) // ps + 1
        torch.manual_seed(seq_len * hd)
        q = _bf(torch.randn(hq, hd)).to(DEV)
        kc = _bf(torch.randn(npg, ps, kh, hd)).to(DEV)
        vc = _bf(torch.randn(npg, ps, kh, hd)).to(DEV)
        perm = torch.randperm(npg, dtype=torch.int32).to(DEV)
        got = ops.attn_decode_paged(q, kc, vc, perm, seq_len)
        want = torch_ref.attn_decode_paged(
            q.cpu(), kc.cpu(), vc.cpu(), perm.cpu(), seq_len
        )
        _assert_close(got, want, atol=2e-2, name=f"decode s={seq_len} hd={hd}")


class TestPrefillAttention:
    @pytest.mark.parametrize("tq", [1, 16, 100, 129, 512])
    def test_mfma_vs_ref_hd128(self, tq):
        hq, kh, hd = 8, 2, 128
        torch.manual_seed(tq)
        q = _bf(torch.randn(tq, hq, hd)).to(DEV)
        k = _bf(torch.randn(tq, kh, hd)).to(DEV)
        v = _bf(torch.randn(tq, kh, hd)).to(DEV)
        got = ops.attn_prefill(q, k, v)
        want = torch_ref.attn_prefill(q.cpu(), k.cpu(), v.cpu())
        _assert_close(got, want, atol=2.5e-2, name=f"prefill mfma tq={tq}")

    @pytest.mark.parametrize("hd", [32, 64])
    def test_simple_vs_ref(self, hd):
        tq, hq, kh = 77, 4, 2
        q = _bf(torch.randn(tq, hq, hd)).to(DEV)
        k = _bf(torch.randn(tq, kh, hd)).to(DEV)
        v = _bf(torch.randn(tq, kh, hd)).to(DEV)
        got = ops.attn_prefill(q, k, v)
        want = torch_ref.attn_prefill(q.cpu(), k.cpu(), v.cpu())
        _assert_close(got, want, atol=2.5e-2, name=f"prefill simple hd={hd}")

    def test_mfma_equals_simple(self):
        from adversarial_spec_amd.ops import _advspec_hip

        tq, hq, kh, hd = 200, 4, 2, 128
        q = _bf(torch.randn(tq, hq, hd)).to(DEV)
        k = _bf(torch.randn(tq, kh, hd)).to(DEV)
        v = _bf(torch.randn(tq, kh, hd)).to(DEV)
        scale = 1.0 / math.sqrt(hd)
        a = _advspec_hip.attn_prefill(q, k, v, scale, True, 0)
        b = _advspec_hip.attn_prefill_simple(q, k, v, scale, True, 0)
        _assert_close(a, b, atol=2e-2, name="mfma vs simple")

    def test_spiked_key_forces_rescale(self):
        """Online-softmax rescale path (guide §5.4 rule 26): one huge key
        late in the sequence forces the running max to jump."""
        tq, hq, kh, hd = 128, 2, 1, 128
        q = _bf(torch.randn(tq, hq, hd) * 0.1).to(DEV)
        k = _bf(torch.randn(tq, kh, hd) * 0.1).to(DEV)
        v = _bf(torch.randn(tq, kh, hd)).to(DEV)
        kf = k.float()
        kf[100] = q.float()[120, 0] * 3.0  # spike aligned with a late query
        k = _bf(kf).to(DEV)
        got = ops.attn_prefill(q, k, v)
        want = torch_ref.attn_prefill(q.cpu(), k.cpu(), v.cpu())
        _assert_close(got, want, atol=2.5e-2, name="prefill spiked")


class TestSampling:
    def test_greedy_is_argmax(self):
        logits = _bf(torch.randn(128256)).to(DEV)
        tok = ops.sample(logits, temperature=0.0)
        assert tok == int(logits.float().argmax().item())

    def test_seeded_deterministic(self):
        logits = _bf(torch.randn(1000)).to(DEV)
        a = ops.sample(logits, temperature=0.7, seed=42)
        b = ops.sample(logits, temperature=0.7, seed=42)
        assert a == b

    def test_distribution_follows_logits(self):
        logits = torch.full((512,), -10.0)
        logits[7] = 5.0
        logits[11] = 4.0
        lb = _bf(logits).to(DEV)
        counts = {}
        for s in range(200):
            t = ops.sample(lb, temperature=1.0, seed=s)
            counts[t] = counts.get(t, 0) + 1
        assert set(counts) <= {7, 11}
        assert counts.get(7, 0) > counts.get(11, 0)

    def test_valid_token_range(self):
        logits = _bf(torch.randn(128256)).to(DEV)
        for s in (1, 2, 3):
            t = ops.sample(logits, temperature=0.7, seed=s)
            assert 0 <= t < 128256


class TestRopeKV:
    def test_fused_matches_rope_then_write(self):
        """rope_kv == rope (q,k) + kv_write(rotated k, v), incl. strided
        qkv views (the fused-QKV slice layout the model passes)."""
        t, hq, kh, hd, ps, npg = 5, 8, 2, 128, 16, 8
        qkv = _bf(torch.randn(t, (hq + 2 * kh) * hd)).to(DEV)
        q = qkv[:, : hq * hd].view(t, hq, hd)
        k = qkv[:, hq * hd : (hq + kh) * hd].view(t, kh, hd)
        v = qkv[:, (hq + kh) * hd :].view(t, kh, hd)
        half = hd // 2
        cos = torch.randn(64, half, device=DEV)
        sin = torch.randn(64, half, device=DEV)
        kc = torch.zeros(npg, ps, kh, hd, dtype=torch.bfloat16, device=DEV)
        vc = torch.zeros_like(kc)
        table = torch.tensor([5, 2, 7, 0, 1, 3, 4, 6], dtype=torch.int32, device=DEV)
        pos0 = 11

        q_ref, k_ref = torch_ref.rope(
            q.float().cpu(), k.float().cpu(), cos.cpu(), sin.cpu(), pos0
        )
        kc_ref = torch.zeros(npg, ps, kh, hd)
        vc_ref = torch.zeros(npg, ps, kh, hd)
        torch_ref.kv_write(
            kc_ref, vc_ref, table.cpu(), pos0,
            k_ref.to(torch.bfloat16).float(), v.float().cpu(),
        )

        q2, k2 = ops.rope_kv(q, k, v, cos, sin, kc, vc, table, pos0)
        _assert_close(q2, q_ref, atol=2e-2, name="rope_kv q")
        _assert_close(k2, k_ref, atol=2e-2, name="rope_kv k")
        _assert_close(kc, kc_ref, atol=2e-2, name="rope_kv k cache")
        assert torch.equal(vc.float().cpu(), vc_ref)

    def test_graph_mode_pos_state(self):
        t, hq, kh, hd, ps, npg = 1, 4, 2, 64, 8, 4
        q = _bf(torch.randn(t, hq, hd)).to(DEV)
        k = _bf(torch.randn(t, kh, hd)).to(DEV)
        v = _bf(torch.randn(t, kh, hd)).to(DEV)
        cos = torch.randn(32, hd // 2, device=DEV)
        sin = torch.randn(32, hd // 2, device=DEV)
        kc = torch.zeros(npg, ps, kh, hd, dtype=torch.bfloat16, device=DEV)
        vc = torch.zeros_like(kc)
        table = torch.arange(npg, dtype=torch.int32, device=DEV)
        pos = 13
        pos_state = torch.tensor([pos], dtype=torch.int32, device=DEV)

        q_want = q.clone(); k_want = k.clone()
        kc2 = kc.clone(); vc2 = vc.clone()
        ops.rope_kv(q_want, k_want, v, cos, sin, kc2, vc2, table, pos)
        q2, k2 = ops.rope_kv(q, k, v, cos, sin, kc, vc, table, 0,
                             pos_state=pos_state)
        assert torch.equal(q2.cpu(), q_want.cpu())
        assert torch.equal(kc.cpu(), kc2.cpu())
        assert torch.equal(vc.cpu(), vc2.cpu())


class TestGemm:
    # weights row-major [out, in]: C = a @ b^T (NT)
    @pytest.mark.parametrize("m", [3, 36, 56, 128, 300])
    def test_vs_torch_fp32(self, m):
        """In-tree MFMA GEMM vs fp32 torch on model-shaped projections."""
        k, n = 512, 1024
        a = _bf(torch.randn(m, k)).to(DEV)
        b = _bf(torch.randn(n, k) * 0.05).to(DEV)
        got = ops.gemm(a, b)
        want = a.float().cpu() @ b.float().cpu().t()
        _assert_close(got, want, atol=3e-2, name=f"gemm m={m}")

    def test_k_edge_and_bigger(self):
        a = _bf(torch.randn(40, 72)).to(DEV)   # K not multiple of 64
        b = _bf(torch.randn(250, 72) * 0.05).to(DEV)  # N not multiple of 128
        got = ops.gemm(a, b)
        want = a.float().cpu() @ b.float().cpu().t()
        _assert_close(got, want, atol=3e-2, name="gemm k-edge")

    def test_full_tile_shape(self):
        a = _bf(torch.randn(256, 2048)).to(DEV)
        b = _bf(torch.randn(1024, 2048) * 0.02).to(DEV)
        got = ops.gemm(a, b)
        want = a.float().cpu() @ b.float().cpu().t()
        _assert_close(got, want, atol=5e-2, name="gemm full-tile")

    def test_deterministic(self):
        a = _bf(torch.randn(77, 2048)).to(DEV)
        b = _bf(torch.randn(4096, 2048) * 0.02).to(DEV)
        x = ops.gemm(a, b)
        for _ in range(3):
            assert torch.equal(ops.gemm(a, b), x)


class TestGemm256:
    """8-phase 256^2 deep-pipelined kernel (gemm256.hip).

    A NEW sync structure (counted vmcnt spanning raw barriers) gets the
    guide's two-lane discipline: refcheck at several shapes + a multi-run
    race screen, and A/B equivalence against the two-barrier 128^2 kernel.
    """

    def _hip(self):
        from adversarial_spec_amd.ops import _load_hip

        return _load_hip()

    @pytest.mark.parametrize("m,n,k", [
        (256, 256, 512),       # single tile, min K
        (512, 512, 1024),      # multi-tile
        (300, 770, 640),       # M and N edges (clamped rows/cols)
        (8192 // 16, 6144 // 4, 4096 // 4),  # model-shaped, scaled
    ])
    def test_vs_torch_fp32(self, m, n, k):
        hip = self._hip()
        a = _bf(torch.randn(m, k)).to(DEV)
        b = _bf(torch.randn(n, k) * 0.05).to(DEV)
        got = hip.gemm_variant(a, b, 256)
        want = a.float().cpu() @ b.float().cpu().t()
        _assert_close(got, want, atol=5e-2, name=f"gemm256 {m}x{n}x{k}")

    def test_matches_128_variant(self):
        hip = self._hip()
        a = _bf(torch.randn(512, 1024)).to(DEV)
        b = _bf(torch.randn(768, 1024) * 0.02).to(DEV)
        g256 = hip.gemm_variant(a, b, 256)
        g128 = hip.gemm_variant(a, b, 128)
        # identical K-order fp32 accumulation -> bitwise equal bf16 outputs
        assert torch.equal(g256, g128)

    def test_race_screen(self):
        """Counted-vmcnt schedules race, not drift: rerun many times at two
        shapes and require bitwise-stable output matching the reference."""
        hip = self._hip()
        for m, n, k in [(256, 512, 512), (512, 256, 1536)]:
            a = _bf(torch.randn(m, k)).to(DEV)
            b = _bf(torch.randn(n, k) * 0.05).to(DEV)
            first = hip.gemm_variant(a, b, 256)
            want = a.float().cpu() @ b.float().cpu().t()
            _assert_close(first, want, atol=5e-2, name=f"race {m}x{n}x{k}")
            for _ in range(10):
                again = hip.gemm_variant(a, b, 256)
                assert torch.equal(again, first), "gemm256 nondeterminism"

    def test_fp8_256_vs_dequant_ref_and_race(self):
        """fp8 8-phase variant (gemm256_fp8.hip): CPU dequant parity at a
        256-dispatch shape + bitwise rerun stability."""
        m, n, k = 512, 768, 640
        x = _bf(torch.randn(m, k)).to(DEV)
        w = _bf(torch.randn(n, k) * 0.05)
        wq, wsc = ops.quantize_fp8_rowwise(w)  # CPU quant -> CPU reference
        got = ops.gemm_fp8(x, wq.to(DEV), wsc.to(DEV))
        want = ops.gemm_fp8(x.cpu(), wq, wsc)
        # atol covers RTNE boundary-ULP differences between the on-device
        # x-quantizer and torch's float8 cast (measured tail: 2/393k
        # elements at ~0.1 with tighter bounds)
        _assert_close(got, want.float(), atol=1.3e-1, rtol=3e-2,
                      name="fp8-256 dequant")
        for _ in range(8):
            again = ops.gemm_fp8(x, wq.to(DEV), wsc.to(DEV))
            assert torch.equal(again, got), "fp8-256 nondeterminism"


class TestGemv:
    @pytest.mark.parametrize("n,k", [(1024, 512), (6144, 4096), (1000, 264)])
    def test_vs_torch_fp32(self, n, k):
        x = _bf(torch.randn(1, k)).to(DEV)
        w = _bf(torch.randn(n, k) * 0.05).to(DEV)
        got = ops.gemv(x, w)
        want = x.float().cpu() @ w.float().cpu().t()
        _assert_close(got, want, atol=3e-2, name=f"gemv {n}x{k}")


class TestFp8:
    def test_quant_kernel_matches_torch_e4m3(self):
        x = _bf(torch.randn(4, 512) * 3).to(DEV)
        q = torch.empty(4, 512, dtype=torch.uint8, device=DEV)
        sc = torch.empty(4, dtype=torch.float32, device=DEV)
        from adversarial_spec_amd.ops import _advspec_hip

        _advspec_hip.quant_fp8(x, q, sc)
        want_q, want_s = ops.quantize_fp8_rowwise(x.cpu())
        assert torch.allclose(sc.cpu(), want_s, rtol=1e-5)
        got = q.cpu().view(torch.float8_e4m3fn).float()
        want = want_q.view(torch.float8_e4m3fn).float()
        # RTNE boundary cases may differ by one ULP step
        frac = (got != want).float().mean().item()
        assert frac < 0.02, f"{frac*100:.2f}% mismatched codes"

    @pytest.mark.parametrize("m", [1, 36, 128, 300])
    def test_gemm_fp8_vs_dequant_ref(self, m):
        k, n = 512, 1024
        x = _bf(torch.randn(m, k)).to(DEV)
        w = _bf(torch.randn(n, k) * 0.05).to(DEV)
        wq, wsc = ops.quantize_fp8_rowwise(w)
        got = ops.gemm_fp8(x, wq, wsc)
        want = ops.gemm_fp8(x.cpu(), wq.cpu(), wsc.cpu())  # CPU dequant ref
        _assert_close(got, want.to(torch.bfloat16), atol=8e-2, rtol=5e-2,
                      name=f"gemm_fp8 m={m}")

    def test_gemv_fp8_vs_gemm_fp8(self):
        k, n = 2048, 4096
        x = _bf(torch.randn(1, k)).to(DEV)
        w = _bf(torch.randn(n, k) * 0.02).to(DEV)
        wq, wsc = ops.quantize_fp8_rowwise(w)
        wq, wsc = wq.to(DEV), wsc.to(DEV)
        x8 = torch.empty(1, k, dtype=torch.uint8, device=DEV)
        xs = torch.empty(1, dtype=torch.float32, device=DEV)
        out = torch.empty(1, n, dtype=torch.bfloat16, device=DEV)
        ops.gemv_fp8(x, wq, wsc, x8, xs, out)
        want = ops.gemm_fp8(x, wq, wsc)
        _assert_close(out, want.float().cpu(), atol=5e-2, rtol=5e-2,
                      name="gemv_fp8")


class TestFp8Engine:
    def test_fp8_generate_close_to_bf16(self):
        """fp8 vs bf16 end-to-end numerics on the SAME random-init weights:
        prefill logits stay close (cosine + top-1 agreement) and the greedy
        decode trajectories match for the first steps (round-1 verdict:
        the old version only checked that both engines emitted tokens)."""
        import torch.nn.functional as F

        from adversarial_spec_amd.engine.local import LocalEngine

        e_bf = LocalEngine({"name": "f8cmp", "arch": "debug-1b"}, device=DEV)
        e_f8 = LocalEngine({"name": "f8cmp", "arch": "debug-1b",
                            "dtype": "fp8"}, device=DEV)

        ids = e_bf.tokenizer.render_chat("s", "fp8 parity prompt")
        toks = torch.tensor(ids, device=DEV, dtype=torch.long)
        c_bf = e_bf.model.new_cache(256)
        c_f8 = e_f8.model.new_cache(256)
        l_bf = e_bf.model.prefill(toks, c_bf).float()
        l_f8 = e_f8.model.prefill(toks, c_f8).float()

        cos = F.cosine_similarity(l_bf.unsqueeze(0), l_f8.unsqueeze(0)).item()
        # rowwise e4m3 on RANDOM weights (std 0.02) carries ~1-2% relative
        # noise per projection; 0.93 cosine bounds the whole debug stack
        assert cos > 0.93, f"fp8 prefill logits cosine {cos}"
        assert int(l_bf.argmax()) == int(l_f8.argmax()), "top-1 diverged"

        # TEACHER-FORCED trajectory: both models consume the bf16 greedy
        # sequence, so per-step logits are comparable without divergence
        # compounding; argmax must agree on most steps
        tok = int(l_bf.argmax().item())
        agree = 0
        for step in range(6):
            lb = e_bf.model.decode_one(tok, c_bf).float()
            lf = e_f8.model.decode_one(tok, c_f8).float()
            c = F.cosine_similarity(lb.unsqueeze(0), lf.unsqueeze(0)).item()
            assert c > 0.90, f"step {step}: logits cosine {c}"
            agree += int(lb.argmax().item()) == int(lf.argmax().item())
            tok = int(lb.argmax().item())
        # random-init logits are near-tied: per-step cosine > 0.90 is the
        # load-bearing check; argmax agreement is a coarse secondary (ULP
        # shifts from kernel-rounding changes flip genuine ties)
        assert agree >= 3, f"fp8 greedy agreed on only {agree}/6 steps"


class TestGemvGateup:
    def test_fused_vs_ref(self):
        """gemv_gateup == silu(x@Wg^T) * (x@Wu^T) on the fused layout."""
        k, f = 1024, 768
        x = _bf(torch.randn(1, k)).to(DEV)
        w = _bf(torch.randn(2 * f, k) * 0.05).to(DEV)
        act = torch.empty(1, f, dtype=torch.bfloat16, device=DEV)
        ops.gemv_gateup(x, w, act)
        gu = x.float().cpu() @ w.float().cpu().t()
        want = torch_ref.swiglu(gu[:, :f], gu[:, f:])
        _assert_close(act, want, atol=3e-2, name="gemv_gateup")

    def test_matches_unfused_ops(self):
        k, f = 4096, 14336  # llama-3-8b decode shape
        x = _bf(torch.randn(1, k)).to(DEV)
        w = _bf(torch.randn(2 * f, k) * 0.02).to(DEV)
        act = torch.empty(1, f, dtype=torch.bfloat16, device=DEV)
        ops.gemv_gateup(x, w, act)
        gu = ops.gemv(x, w)
        want = ops.swiglu(gu[:, :f], gu[:, f:])
        _assert_close(act, want.float().cpu(), atol=3e-2, name="fused vs ops")


class TestGemvNormResFusion:
    """Norm-prologue / residual-epilogue GEMV fusion (decode launch-count
    reduction): each fused kernel vs the plain fp32 torch reference."""

    EPS = 1e-5

    def _norm_ref(self, x, wln):
        xf = x.float().cpu()
        rms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + self.EPS)
        return xf * rms * wln.float().cpu()

    def test_gemv_norm_w32(self):
        k, n = 4096, 6144  # qkv shape (w32 kernel)
        x = _bf(torch.randn(1, k)).to(DEV)
        wln = _bf(torch.rand(k) + 0.5).to(DEV)
        w = _bf(torch.randn(n, k) * 0.02).to(DEV)
        y = ops.gemv_norm(x, wln, w, self.EPS)
        want = self._norm_ref(x, wln) @ w.float().cpu().t()
        _assert_close(y, want, atol=8e-2, name="gemv_norm w32")

    def test_gemv_norm_large_n(self):
        k, n = 512, 16384  # 16-lane kernel path (N > 8192, lm_head-like)
        x = _bf(torch.randn(1, k)).to(DEV)
        wln = _bf(torch.rand(k) + 0.5).to(DEV)
        w = _bf(torch.randn(n, k) * 0.05).to(DEV)
        y = ops.gemv_norm(x, wln, w, self.EPS)
        want = self._norm_ref(x, wln) @ w.float().cpu().t()
        _assert_close(y, want, atol=8e-2, name="gemv_norm 16-lane")

    def test_gemv_res_inplace(self):
        k, n = 4096, 4096  # o/down shape
        x = _bf(torch.randn(1, k)).to(DEV)
        w = _bf(torch.randn(n, k) * 0.02).to(DEV)
        resid = _bf(torch.randn(1, n)).to(DEV)
        want = resid.float().cpu() + x.float().cpu() @ w.float().cpu().t()
        ops.gemv_res(x, w, resid)
        _assert_close(resid, want, atol=8e-2, name="gemv_res")

    def test_gemv_gateup_norm(self):
        k, f = 4096, 14336
        x = _bf(torch.randn(1, k)).to(DEV)
        wln = _bf(torch.rand(k) + 0.5).to(DEV)
        w = _bf(torch.randn(2 * f, k) * 0.02).to(DEV)
        act = torch.empty(1, f, dtype=torch.bfloat16, device=DEV)
        ops.gemv_gateup_norm(x, wln, w, self.EPS, act)
        gu = self._norm_ref(x, wln) @ w.float().cpu().t()
        want = torch_ref.swiglu(gu[:, :f], gu[:, f:])
        _assert_close(act, want, atol=5e-2, name="gemv_gateup_norm")

    def test_fp8_quant_norm(self):
        """quant_norm_fp8 == rmsnorm then rowwise-e4m3 quantize."""
        k = 4096
        x = _bf(torch.randn(1, k)).to(DEV)
        wln = _bf(torch.rand(k) + 0.5).to(DEV)
        x8 = torch.empty(1, k, dtype=torch.uint8, device=DEV)
        xs = torch.empty(1, dtype=torch.float32, device=DEV)
        ops.quant_norm_fp8(x, wln, x8, xs, self.EPS)
        deq = (x8.view(torch.float8_e4m3fn).float() * xs).cpu()
        want = self._norm_ref(x, wln)
        # e4m3 has ~2^-3 relative mantissa step at the top of each binade
        err = (deq - want).abs().max().item()
        scale = want.abs().max().item()
        assert err <= scale * 0.08, f"quant_norm err {err} vs amax {scale}"

    def test_fp8_fused_gemv_chain(self):
        """quant_norm -> gemv_fp8_q / gemv_fp8_gateup / gemv_fp8_res vs the
        fp32 reference of the same quantized weights."""
        k, n, f = 2048, 1024, 1536
        x = _bf(torch.randn(1, k)).to(DEV)
        wln = _bf(torch.rand(k) + 0.5).to(DEV)
        w = _bf(torch.randn(n, k) * 0.02).to(DEV)
        wgu = _bf(torch.randn(2 * f, k) * 0.02).to(DEV)
        wq, wsc = ops.quantize_fp8_rowwise(w)
        gq, gsc = ops.quantize_fp8_rowwise(wgu)
        x8 = torch.empty(1, k, dtype=torch.uint8, device=DEV)
        xs = torch.empty(1, dtype=torch.float32, device=DEV)
        ops.quant_norm_fp8(x, wln, x8, xs, self.EPS)
        xref = (x8.view(torch.float8_e4m3fn).float() * xs).cpu()

        out = torch.empty(1, n, dtype=torch.bfloat16, device=DEV)
        ops.gemv_fp8_q(x8, xs, wq, wsc, out)
        want = xref @ ops.dequantize_fp8(wq.cpu(), wsc.cpu()).t()
        _assert_close(out, want, atol=1.5e-1, name="gemv_fp8_q")

        act = torch.empty(1, f, dtype=torch.bfloat16, device=DEV)
        ops.gemv_fp8_gateup(x8, xs, gq, gsc, act)
        gu = xref @ ops.dequantize_fp8(gq.cpu(), gsc.cpu()).t()
        want_act = torch_ref.swiglu(gu[:, :f], gu[:, f:])
        _assert_close(act, want_act, atol=1e-1, name="gemv_fp8_gateup")

        resid = _bf(torch.randn(1, n)).to(DEV)
        xin = _bf(torch.randn(1, k)).to(DEV)
        xin8 = (xin.float() / (xin.float().abs().amax() / 448.0)
                ).to(torch.float8_e4m3fn).float().cpu() * (
            xin.float().abs().amax().cpu() / 448.0)
        want_res = resid.float().cpu() + \
            xin8 @ ops.dequantize_fp8(wq.cpu(), wsc.cpu()).t()
        ops.gemv_fp8_res(xin, wq, wsc, x8, xs, resid)
        _assert_close(resid, want_res, atol=1.5e-1, name="gemv_fp8_res")

    def test_fp8_lds_fused_kernels(self):
        """Single-launch LDS-staged fp8 GEMVs (norm / res / gateup_norm)
        vs the fp32 reference of the same quantized weights."""
        k, n, f = 2048, 1024, 1536
        x = _bf(torch.randn(1, k)).to(DEV)
        wln = _bf(torch.rand(k) + 0.5).to(DEV)
        w = _bf(torch.randn(n, k) * 0.02).to(DEV)
        wgu = _bf(torch.randn(2 * f, k) * 0.02).to(DEV)
        wq, wsc = ops.quantize_fp8_rowwise(w)
        gq, gsc = ops.quantize_fp8_rowwise(wgu)

        normed = self._norm_ref(x, wln)
        s = float(normed.abs().amax()) / 448.0
        n8 = (normed / s).to(torch.float8_e4m3fn).float() * s
        wd = ops.dequantize_fp8(wq.cpu(), wsc.cpu())
        gd = ops.dequantize_fp8(gq.cpu(), gsc.cpu())

        out = torch.empty(1, n, dtype=torch.bfloat16, device=DEV)
        ops.gemv_fp8_norm(x, wln, wq, wsc, self.EPS, out)
        _assert_close(out, n8 @ wd.t(), atol=1.5e-1, name="gemv_fp8_norm")

        act = torch.empty(1, f, dtype=torch.bfloat16, device=DEV)
        ops.gemv_fp8_gateup_norm(x, wln, gq, gsc, self.EPS, act)
        gu = n8 @ gd.t()
        _assert_close(act, torch_ref.swiglu(gu[:, :f], gu[:, f:]),
                      atol=1e-1, name="gemv_fp8_gateup_norm")

        resid = _bf(torch.randn(1, n)).to(DEV)
        xin = _bf(torch.randn(1, k)).to(DEV)
        si = float(xin.float().abs().amax()) / 448.0
        xi8 = (xin.float().cpu() / si).to(torch.float8_e4m3fn).float() * si
        want = resid.float().cpu() + xi8 @ wd.t()
        ops.gemv_fp8_resl(xin, wq, wsc, resid)
        _assert_close(resid, want, atol=1.5e-1, name="gemv_fp8_resl")

    def test_fused_decode_step_matches_unfused_forward(self):
        """decode_step_ws (fused GEMV path) vs decode_one (unfused kernel
        sequence) on the same prefilled cache: same logits direction."""
        from adversarial_spec_amd.models import LlamaModel
        from adversarial_spec_amd.models.config import LlamaConfig

        cfg = LlamaConfig(
            name="fuse-t", dim=512, n_layers=3, n_heads=4, n_kv_heads=2,
            ffn_dim=1024, vocab_size=1024, max_seq_len=512,
            rope_theta=10000.0,
        )
        m = LlamaModel(cfg, device=DEV, dtype=torch.bfloat16, seed=3).init_random()
        toks = torch.arange(1, 33, device=DEV)
        c1 = m.new_cache(256)
        m.prefill(toks[:-1], c1)
        ref = m.decode_one(int(toks[-1]), c1).float()

        c2 = m.new_cache(256)
        m.prefill(toks[:-1], c2)
        W = m.new_decode_ws()
        W.tok_long.fill_(int(toks[-1]))
        pos_state = torch.tensor([c2.seq_len], dtype=torch.int32, device=DEV)
        got = m.decode_step_ws(c2, pos_state, 256, W)[0].float()
        cos = torch.nn.functional.cosine_similarity(
            ref.unsqueeze(0), got.unsqueeze(0)
        ).item()
        assert cos > 0.995, f"fused/unfused decode cosine {cos}"
        assert int(ref.argmax()) == int(got.argmax())

    def test_fused_fp8_decode_step_matches_unfused_forward(self):
        """fp8 decode_step_ws (quant_norm/res/gateup fusion) vs the
        unfused fp8 decode_one kernel sequence."""
        from adversarial_spec_amd.models import LlamaModel
        from adversarial_spec_amd.models.config import LlamaConfig

        cfg = LlamaConfig(
            name="fuse-f8", dim=512, n_layers=3, n_heads=4, n_kv_heads=2,
            ffn_dim=1024, vocab_size=1024, max_seq_len=512,
            rope_theta=10000.0,
        )
        m = LlamaModel(cfg, device=DEV, dtype=torch.bfloat16, seed=5)
        m.init_random().quantize_fp8()
        toks = torch.arange(1, 33, device=DEV)
        c1 = m.new_cache(256)
        m.prefill(toks[:-1], c1)
        ref = m.decode_one(int(toks[-1]), c1).float()

        c2 = m.new_cache(256)
        m.prefill(toks[:-1], c2)
        W = m.new_decode_ws()
        W.tok_long.fill_(int(toks[-1]))
        pos_state = torch.tensor([c2.seq_len], dtype=torch.int32, device=DEV)
        got = m.decode_step_ws(c2, pos_state, 256, W)[0].float()
        cos = torch.nn.functional.cosine_similarity(
            ref.unsqueeze(0), got.unsqueeze(0)
        ).item()
        assert cos > 0.99, f"fp8 fused/unfused decode cosine {cos}"
        assert int(ref.argmax()) == int(got.argmax())
