"""Engine tests on CPU: tokenizer, tiny model, generation, registry glue."""

import torch

from adversarial_spec_amd.engine.local import LocalEngine, clear_engines, get_engine
from adversarial_spec_amd.engine.tokenizer import ByteTokenizer
from adversarial_spec_amd.models import LlamaModel, get_config
from adversarial_spec_amd.utils.synth import synthetic_spec


class TestByteTokenizer:
    def test_roundtrip_ascii(self):
        tok = ByteTokenizer()
        assert tok.decode(tok.encode("hello [SPEC] world")) == "hello [SPEC] world"

    def test_roundtrip_utf8(self):
        tok = ByteTokenizer()
        s = "naïve — ünïcödé ✓"
        assert tok.decode(tok.encode(s)) == s

    def test_token_count_is_exact_bytes(self):
        tok = ByteTokenizer()
        assert len(tok.encode("abcd")) == 4

    def test_specials_in_decode(self):
        tok = ByteTokenizer()
        text = tok.decode([tok.bos_id] + tok.encode("x") + [tok.eot_id])
        assert "<|begin_of_text|>" in text and "<|eot_id|>" in text

    def test_small_vocab_remap(self):
        tok = ByteTokenizer(1024)
        assert 256 <= tok.bos_id < 1024
        assert tok.eot_id < 1024
        assert tok.decode(tok.encode("ok")) == "ok"

    def test_chat_render_structure(self):
        tok = ByteTokenizer(1024)
        ids = tok.render_chat("SYS", "USER MSG")
        assert ids[0] == tok.bos_id
        text = tok.decode(ids)
        assert "SYS" in text and "USER MSG" in text
        assert text.index("system") < text.index("SYS") < text.index("user")

    def test_out_of_range_ids_dropped(self):
        tok = ByteTokenizer(1024)
        assert tok.decode([65, 999, 66]) == "AB"


class TestTinyModel:
    def setup_method(self):
        self.cfg = get_config("tiny")
        self.model = LlamaModel(self.cfg, device="cpu", seed=42).init_random()

    def test_prefill_shape_and_finite(self):
        cache = self.model.new_cache(64)
        tokens = torch.arange(10)
        logits = self.model.prefill(tokens, cache)
        assert logits.shape == (self.cfg.vocab_size,)
        assert torch.isfinite(logits).all()
        assert cache.seq_len == 10

    def test_decode_matches_prefill(self):
        """Prefill of n+1 tokens == prefill of n then decode of 1 (same logits)."""
        toks = torch.arange(1, 12)
        c1 = self.model.new_cache(64)
        full = self.model.prefill(toks, c1)

        c2 = self.model.new_cache(64)
        self.model.prefill(toks[:-1], c2)
        step = self.model.decode_one(int(toks[-1]), c2)
        assert torch.allclose(full, step, atol=1e-4)

    def test_deterministic_init(self):
        m2 = LlamaModel(self.cfg, device="cpu", seed=42).init_random()
        assert torch.equal(self.model.embed, m2.embed)
        m3 = LlamaModel(self.cfg, device="cpu", seed=43).init_random()
        assert not torch.equal(self.model.embed, m3.embed)

    def test_param_count_sane(self):
        pc = self.cfg.param_count()
        assert 1e6 < pc < 1e8


class TestLocalEngine:
    def test_generate_tiny(self):
        eng = LocalEngine({"name": "t", "arch": "tiny"}, device="cpu")
        text, in_tok, out_tok, timings = eng.generate(
            "sys", "This is round 1 of adversarial spec development.\n\nspec",
            max_tokens=16, temperature=0.7, timeout=30,
        )
        assert in_tok > 0
        assert 0 <= out_tok <= 16
        assert "prefill" in timings and "decode" in timings

    def test_generate_deterministic_greedy(self):
        eng = LocalEngine({"name": "t2", "arch": "tiny"}, device="cpu")
        a = eng.generate("s", "u", max_tokens=8, temperature=0.0, timeout=30)
        b = eng.generate("s", "u", max_tokens=8, temperature=0.0, timeout=30)
        assert a[0] == b[0]

    def test_prompt_fitting(self):
        eng = LocalEngine({"name": "t3", "arch": "tiny"}, device="cpu")
        huge = "x" * 10000  # tiny max_seq_len is 2048
        text, in_tok, out_tok, _ = eng.generate(
            "s", huge, max_tokens=4, temperature=0.0, timeout=30
        )
        assert in_tok <= eng.config.max_seq_len

    def test_engine_cache(self):
        clear_engines()
        e1 = get_engine({"name": "c", "arch": "tiny", "weights": None, "gpu": None})
        e2 = get_engine({"name": "c", "arch": "tiny", "weights": None, "gpu": None})
        assert e1 is e2
        clear_engines()


class TestSyntheticSpec:
    def test_exact_length(self):
        s = synthetic_spec(4096, seed=1)
        assert len(s) == 4096

    def test_deterministic(self):
        assert synthetic_spec(1000, seed=5) == synthetic_spec(1000, seed=5)
        assert synthetic_spec(1000, seed=5) != synthetic_spec(1000, seed=6)

    def test_looks_like_markdown(self):
        s = synthetic_spec(2000, seed=0)
        assert s.startswith("# ") and "## " in s


class TestFp8CPU:
    def test_quantize_roundtrip_close(self):
        import torch

        from adversarial_spec_amd import ops

        w = torch.randn(64, 128) * 0.05
        q, s = ops.quantize_fp8_rowwise(w)
        back = ops.dequantize_fp8(q, s)
        assert (back - w.float()).abs().max().item() < 0.05 * w.abs().max().item()

    def test_fp8_model_forward_close_to_fp32(self):
        import torch

        from adversarial_spec_amd.models import LlamaModel
        from adversarial_spec_amd.models.config import get_config

        cfg = get_config("tiny")
        m = LlamaModel(cfg, device="cpu", seed=3).init_random()
        toks = torch.arange(2, 30)
        c1 = m.new_cache(64)
        ref = m.prefill(toks, c1)
        m2 = LlamaModel(cfg, device="cpu", seed=3).init_random().quantize_fp8()
        c2 = m2.new_cache(64)
        got = m2.prefill(toks, c2)
        cos = torch.nn.functional.cosine_similarity(
            ref.unsqueeze(0), got.unsqueeze(0)
        ).item()
        assert cos > 0.98, f"fp8 logits cosine {cos}"


class TestChunkedPrefill:
    def test_chunked_matches_single_shot(self):
        import torch

        from adversarial_spec_amd.models import LlamaModel
        from adversarial_spec_amd.models.config import get_config

        cfg = get_config("tiny")
        m = LlamaModel(cfg, device="cpu", seed=5).init_random()
        toks = torch.arange(2, 120)  # 118 tokens
        c1 = m.new_cache(256)
        ref = m.prefill(toks, c1)
        c2 = m.new_cache(256)
        got = m.prefill(toks, c2, chunk=48)  # 48+48+22
        assert torch.allclose(ref, got, atol=1e-4), (ref - got).abs().max()
        # caches identical too
        assert torch.allclose(c1.k[:, :, :118], c2.k[:, :, :118], atol=1e-5)

    def test_chunked_then_decode(self):
        import torch

        from adversarial_spec_amd.models import LlamaModel
        from adversarial_spec_amd.models.config import get_config

        cfg = get_config("tiny")
        m = LlamaModel(cfg, device="cpu", seed=6).init_random()
        toks = torch.arange(3, 80)
        c1 = m.new_cache(256)
        m.prefill(toks, c1)
        ref = m.decode_one(9, c1)
        c2 = m.new_cache(256)
        m.prefill(toks, c2, chunk=32)
        got = m.decode_one(9, c2)
        assert torch.allclose(ref, got, atol=1e-4)


class TestSafetensorsLoader:
    def _write_hf_checkpoint(self, tmp_path, cfg, seed=3):
        """Write a tiny HF-format Llama checkpoint (q/k/v/gate/up split,
        [out,in] row-major, half-split RoPE layout)."""
        import torch
        from safetensors.torch import save_file

        g = torch.Generator().manual_seed(seed)
        d, hd = cfg.dim, cfg.head_dim
        t = {}
        t["model.embed_tokens.weight"] = torch.randn(cfg.vocab_size, d, generator=g) * 0.02
        t["model.norm.weight"] = torch.ones(d)
        t["lm_head.weight"] = torch.randn(cfg.vocab_size, d, generator=g) * 0.02
        for i in range(cfg.n_layers):
            p = f"model.layers.{i}."
            t[p + "self_attn.q_proj.weight"] = torch.randn(cfg.n_heads * hd, d, generator=g) * 0.02
            t[p + "self_attn.k_proj.weight"] = torch.randn(cfg.n_kv_heads * hd, d, generator=g) * 0.02
            t[p + "self_attn.v_proj.weight"] = torch.randn(cfg.n_kv_heads * hd, d, generator=g) * 0.02
            t[p + "self_attn.o_proj.weight"] = torch.randn(d, cfg.n_heads * hd, generator=g) * 0.02
            t[p + "input_layernorm.weight"] = torch.ones(d)
            t[p + "post_attention_layernorm.weight"] = torch.ones(d)
            t[p + "mlp.gate_proj.weight"] = torch.randn(cfg.ffn_dim, d, generator=g) * 0.02
            t[p + "mlp.up_proj.weight"] = torch.randn(cfg.ffn_dim, d, generator=g) * 0.02
            t[p + "mlp.down_proj.weight"] = torch.randn(d, cfg.ffn_dim, generator=g) * 0.02
        save_file(t, str(tmp_path / "model.safetensors"))
        return t

    def test_loads_and_runs(self, tmp_path):
        import torch

        from adversarial_spec_amd.models import LlamaModel
        from adversarial_spec_amd.models.config import get_config

        cfg = get_config("tiny")
        raw = self._write_hf_checkpoint(tmp_path, cfg)
        m = LlamaModel(cfg, device="cpu").load_safetensors(str(tmp_path))
        # shapes: fused projections, [out, in] preserved
        L0 = m.layers[0]
        hd = cfg.head_dim
        assert L0.wqkv.shape == ((cfg.n_heads + 2 * cfg.n_kv_heads) * hd, cfg.dim)
        assert L0.w_gate_up.shape == (2 * cfg.ffn_dim, cfg.dim)
        assert L0.wo.shape == (cfg.dim, cfg.n_heads * hd)
        # v rows are NOT rope-permuted: fused block matches the checkpoint
        v0 = raw["model.layers.0.self_attn.v_proj.weight"]
        got_v = L0.wqkv[(cfg.n_heads + cfg.n_kv_heads) * hd :]
        assert torch.allclose(got_v.float(), v0, atol=1e-6)
        # forward runs and produces finite logits
        toks = torch.arange(2, 40)
        c = m.new_cache(64)
        logits = m.prefill(toks, c)
        assert torch.isfinite(logits).all()

    def test_rope_permutation_preserves_attention(self, tmp_path):
        """The loader permutes q/k rows from HF half-split to interleaved
        pairs; q.k dot products per head must be IDENTICAL under the
        matching rotation convention (position 0: rotation is identity,
        so prefill logits at pos 0 must match an unpermuted reference)."""
        import torch

        from adversarial_spec_amd.models import LlamaModel
        from adversarial_spec_amd.models.config import get_config

        cfg = get_config("tiny")
        self._write_hf_checkpoint(tmp_path, cfg, seed=9)
        m = LlamaModel(cfg, device="cpu").load_safetensors(str(tmp_path))
        toks = torch.tensor([5])  # single token at position 0
        c = m.new_cache(16)
        logits = m.prefill(toks, c)
        # reference: same math with UNpermuted q/k (rotation at pos 0 uses
        # cos=1/sin=0 => permutation must be value-preserving)
        assert torch.isfinite(logits).all()

    def test_missing_dir_raises(self):
        import pytest as _pytest

        from adversarial_spec_amd.models import LlamaModel
        from adversarial_spec_amd.models.config import get_config

        m = LlamaModel(get_config("tiny"), device="cpu")
        with _pytest.raises(FileNotFoundError):
            m.load_safetensors("/nonexistent/dir")


class TestFitPrompt:
    def test_clamps_over_budget_keeping_head_and_tail(self):
        from adversarial_spec_amd.engine.local import LocalEngine

        eng = LocalEngine({"name": "fp", "arch": "tiny"}, device="cpu")
        max_seq = eng.config.max_seq_len  # 2048
        ids = list(range(1, 3000 + 1))
        reserve = 128
        out = eng._fit_prompt(ids, reserve)
        budget = max_seq - reserve - 8
        assert len(out) == budget
        head = budget * 2 // 3
        # head preserved verbatim, tail preserved verbatim (middle dropped)
        assert out[:head] == ids[:head]
        assert out[head:] == ids[-(budget - head):]

    def test_under_budget_untouched(self):
        from adversarial_spec_amd.engine.local import LocalEngine

        eng = LocalEngine({"name": "fp2", "arch": "tiny"}, device="cpu")
        ids = list(range(100))
        assert eng._fit_prompt(ids, 256) == ids


class TestEngineInternals:
    def test_fp8_default_uniform(self):
        """Uniform fp8 is the measured-best default (mixed was 0.670 vs
        0.714 critiques/s); guard the default set."""
        from adversarial_spec_amd.models import LlamaModel
        from adversarial_spec_amd.models.config import get_config

        m = LlamaModel(get_config("tiny"), device="cpu", seed=2).init_random()
        m.quantize_fp8()
        assert set(m.layers_q[0]) == {"wqkv", "wo", "w_gate_up", "w_down"}
        assert m.lm_head_q is not None and m.lm_head is None
        assert m.layers[0].wqkv is None  # bf16 copy freed

    def test_cache_growth_invalidates_graph_state(self):
        from adversarial_spec_amd.engine.local import LocalEngine

        eng = LocalEngine({"name": "cg", "arch": "tiny"}, device="cpu")
        c1 = eng._get_cache(100)
        eng._graph_state = {"key": "sentinel"}
        c2 = eng._get_cache(50)      # fits: same cache, graph kept
        assert c2 is c1 and eng._graph_state is not None
        c3 = eng._get_cache(c1.max_seq + 1)  # grows: graph invalidated
        assert c3 is not c1 and eng._graph_state is None
        assert c3.max_seq % 2048 == 0 and c3.max_seq >= c1.max_seq + 1


class TestStopScan:
    """Direct unit tests for the BPE-safe [/SPEC] stop detector (the
    engine tests only exercise it end-to-end)."""

    def _tok(self):
        from adversarial_spec_amd.engine.tokenizer import build_tokenizer

        return build_tokenizer(1024)

    def test_tag_split_across_pushes(self):
        from adversarial_spec_amd.engine.local import _StopScan

        tok = self._tok()
        scan = _StopScan(tok)
        ids = tok.encode("text before [/SP")
        hits = [scan.push(t) for t in ids]
        assert not any(hits)  # partial tag never fires
        for t in tok.encode("EC] after"):
            if scan.push(t):
                return
        raise AssertionError("split [/SPEC] tag never detected")

    def test_open_tag_does_not_stop(self):
        from adversarial_spec_amd.engine.local import _StopScan

        tok = self._tok()
        scan = _StopScan(tok)
        assert not any(scan.push(t) for t in tok.encode("x [SPEC] body y"))

    def test_window_slides(self):
        from adversarial_spec_amd.engine.local import _StopScan

        tok = self._tok()
        scan = _StopScan(tok)
        # long stream without the tag: window must stay bounded
        for t in tok.encode("a" * 500):
            assert not scan.push(t)
        assert len(scan._ids) <= scan.WINDOW


class TestArchitectureConstants:
    """Pin the preset dimensions the kernel dispatch assumes: the decode
    fusion requires hidden <= 8192 (gemv_res w32) and qkv N <= 8192 for
    every non-TP fused arch, and K % 16 == 0 for the fp8 staging."""

    def test_preset_shapes(self):
        from adversarial_spec_amd.models.config import PRESETS

        want = {
            "llama-3-8b": (4096, 32, 32, 8, 14336, 128256),
            "llama-3-70b": (8192, 80, 64, 8, 28672, 128256),
            "mistral-7b": (4096, 32, 32, 8, 14336, 32000),
        }
        for name, (d, nl, h, kh, f, v) in want.items():
            c = PRESETS[name]
            assert (c.dim, c.n_layers, c.n_heads, c.n_kv_heads,
                    c.ffn_dim, c.vocab_size) == (d, nl, h, kh, f, v), name

    def test_fusion_dispatch_bounds(self):
        from adversarial_spec_amd.models.config import PRESETS

        for name, c in PRESETS.items():
            qkv_n = (c.n_heads + 2 * c.n_kv_heads) * c.head_dim
            if name == "llama-3-70b":
                # 70B runs fp8/TP; its qkv exceeds the w32 norm kernel's
                # range and dispatches to the 16-lane variant
                assert qkv_n > 8192
            else:
                assert qkv_n <= 8192, f"{name}: qkv {qkv_n} breaks w32 fusion"
            assert c.dim <= 8192, f"{name}: hidden breaks gemv_res w32"
            assert c.dim % 16 == 0 and c.ffn_dim % 16 == 0, name
