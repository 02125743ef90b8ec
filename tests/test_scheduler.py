"""Round scheduler tests: fan-out, retries, fault isolation, prompt assembly."""

from unittest.mock import patch

from adversarial_spec_amd.engine.backend import StubBackend, is_o_series_model
from adversarial_spec_amd.engine.scheduler import (
    build_user_message,
    call_models_parallel,
    call_single_model,
    load_context_files,
)


class TestOSeries:
    def test_o1(self):
        assert is_o_series_model("o1")
        assert is_o_series_model("o1-mini")
        assert is_o_series_model("openai/o1-preview")
        assert is_o_series_model("O1")

    def test_not_o_series(self):
        assert not is_o_series_model("gpt-4o")
        assert not is_o_series_model("local/llama-3-8b")


class TestPromptAssembly:
    def test_review_basic(self):
        msg = build_user_message("SPEC", 1, "tech")
        assert "round 1" in msg and "SPEC" in msg
        assert "Technical Specification" in msg

    def test_press_template(self):
        msg = build_user_message("SPEC", 2, "tech", press=True)
        assert "previously signalled agreement" in msg

    def test_focus_known(self):
        msg = build_user_message("S", 1, "tech", focus="security")
        assert "CRITICAL FOCUS: SECURITY" in msg

    def test_focus_custom(self):
        msg = build_user_message("S", 1, "tech", focus="compliance")
        assert "CRITICAL FOCUS: COMPLIANCE" in msg

    def test_preserve_intent_prefixes_focus(self):
        msg = build_user_message("S", 1, "tech", focus="security", preserve_intent=True)
        assert msg.index("PRESERVE ORIGINAL INTENT") < msg.index("CRITICAL FOCUS: SECURITY")

    def test_context_included(self):
        msg = build_user_message("S", 1, "tech", context="## Additional Context\nCTX")
        assert "CTX" in msg

    def test_load_context_files(self, tmp_path):
        f = tmp_path / "api.md"
        f.write_text("api details")
        out = load_context_files([str(f)])
        assert "api details" in out and str(f) in out

    def test_load_context_missing_file(self):
        out = load_context_files(["/does/not/exist.md"])
        assert "Error loading file" in out

    def test_load_context_empty(self):
        assert load_context_files([]) == ""


class TestCallSingleModel:
    def test_stub_critique_parsed(self, fresh_cost_tracker):
        r = call_single_model("stub/critique", "SPEC", 1, "tech")
        assert not r.error
        assert not r.agreed
        assert r.spec is not None
        assert r.input_tokens > 0 and r.output_tokens > 0
        assert r.cost >= 0

    def test_stub_agree(self, fresh_cost_tracker):
        r = call_single_model("stub/agree", "SPEC", 1, "tech")
        assert r.agreed and r.spec

    def test_error_isolated(self, fresh_cost_tracker):
        with patch("adversarial_spec_amd.engine.scheduler.time.sleep"):
            r = call_single_model("stub/error", "SPEC", 1, "tech")
        assert r.error and "simulated failure" in r.error
        assert r.response == ""

    def test_retry_succeeds_after_flaky(self, fresh_cost_tracker):
        StubBackend.reset_flaky()
        sleeps = []
        with patch("adversarial_spec_amd.engine.scheduler.time.sleep", sleeps.append):
            r = call_single_model("stub/flaky", "SPEC", 1, "tech")
        assert not r.error
        # exponential backoff 1s then 2s (reference: models.py:46-47)
        assert sleeps == [1.0, 2.0]

    def test_round_2_auto_agrees(self, fresh_cost_tracker):
        r = call_single_model("stub/auto", "SPEC", 2, "tech")
        assert r.agreed


class TestCallModelsParallel:
    def test_results_in_model_order(self, fresh_cost_tracker):
        models = ["stub/agree", "stub/critique", "stub/agree"]
        results = call_models_parallel(models, "SPEC", 1, "tech")
        assert [r.model for r in results] == models

    def test_mixed_round(self, fresh_cost_tracker):
        with patch("adversarial_spec_amd.engine.scheduler.time.sleep"):
            results = call_models_parallel(
                ["stub/agree", "stub/error", "stub/critique"], "SPEC", 1, "tech"
            )
        assert results[0].agreed
        assert results[1].error
        assert results[2].spec

    def test_env_forces_stub(self, fresh_cost_tracker, monkeypatch):
        monkeypatch.setenv("ADVSPEC_BACKEND", "stub")
        results = call_models_parallel(["gpt-4o"], "SPEC", 2, "tech")
        assert not results[0].error
        assert results[0].agreed


class TestMutationKillers:
    """Killers for tools/mutation_check.py survivors (scheduler)."""

    def test_retry_sleeps_exactly_max_retries_minus_one(self, monkeypatch):
        """All attempts fail -> MAX_RETRIES calls, MAX_RETRIES-1 backoffs of
        1s/2s/4s (reference models.py:46-47, 664)."""
        from adversarial_spec_amd.engine import scheduler
        from adversarial_spec_amd.protocol import MAX_RETRIES

        sleeps = []
        monkeypatch.setattr(scheduler.time, "sleep", sleeps.append)
        calls = {"n": 0}

        class Boom:
            def generate(self, *a, **k):
                calls["n"] += 1
                raise RuntimeError("always fails")

        monkeypatch.setattr(scheduler, "get_backend",
                            lambda m, **kw: Boom())
        r = scheduler.call_single_model("stub/x", "spec", 1, "tech")
        assert r.error is not None
        assert calls["n"] == MAX_RETRIES
        assert sleeps == [1.0 * (2**i) for i in range(MAX_RETRIES - 1)]

    def test_assign_devices_round_robin_increment(self, monkeypatch):
        from adversarial_spec_amd.engine import scheduler

        class FakeCuda:
            @staticmethod
            def is_available():
                return True

            @staticmethod
            def device_count():
                return 2

        import torch

        monkeypatch.setattr(torch, "cuda", FakeCuda)
        out = scheduler._assign_devices(
            ["local/a", "gpt-4o", "local/b", "local/c"]
        )
        # local opponents pinned 0,1,0; remote unpinned
        assert out == ["cuda:0", None, "cuda:1", "cuda:0"]

    def test_default_press_uses_review_template(self, monkeypatch):
        """Calling WITHOUT press must produce the REVIEW prompt (the PRESS
        template demands section listing; defaults flipping to True would
        swap every round-1 prompt)."""
        from adversarial_spec_amd.engine import scheduler

        seen = {}

        class Cap:
            def generate(self, system, user, **k):
                seen["user"] = user
                return "[AGREE]", 1, 1

        monkeypatch.setattr(scheduler, "get_backend",
                            lambda m, **kw: Cap())
        scheduler.call_single_model("stub/x", "MYSPEC", 1, "tech")
        from adversarial_spec_amd.prompts import PRESS_PROMPT_TEMPLATE

        assert "previously signalled agreement" in PRESS_PROMPT_TEMPLATE
        assert "previously signalled agreement" not in seen["user"]
        assert "MYSPEC" in seen["user"]

    def test_assign_devices_no_gpu_leaves_unpinned(self, monkeypatch):
        """ngpu == 0 must take the unpinned branch (a >= mutant divides by
        zero in `local_idx % ngpu`)."""
        from adversarial_spec_amd.engine import scheduler

        class NoCuda:
            @staticmethod
            def is_available():
                return False

            @staticmethod
            def device_count():
                return 0

        import torch

        monkeypatch.setattr(torch, "cuda", NoCuda)
        assert scheduler._assign_devices(["local/a", "local/b"]) == [None, None]

    def test_single_model_flag_defaults_reach_backend(self, monkeypatch):
        """codex_search / bedrock_mode / preserve_intent default to OFF and
        must arrive at the backend factory and prompt that way."""
        from adversarial_spec_amd.engine import scheduler

        seen = {}

        class Cap:
            def generate(self, system, user, **k):
                seen["user"] = user
                return "[AGREE]", 1, 1

        def capture_backend(model, **kw):
            seen["backend_kwargs"] = kw
            return Cap()

        monkeypatch.setattr(scheduler, "get_backend", capture_backend)
        scheduler.call_single_model("stub/x", "SPEC", 1, "tech")
        assert seen["backend_kwargs"]["codex_search"] is False
        assert seen["backend_kwargs"]["bedrock_mode"] is False
        assert "ERRORS" not in seen["user"]  # preserve-intent taxonomy absent

    def test_parallel_flag_defaults_reach_backend(self, monkeypatch):
        from adversarial_spec_amd.engine import scheduler

        seen = {}

        class Cap:
            def generate(self, system, user, **k):
                seen["user"] = user
                return "[AGREE]", 1, 1

        def capture_backend(model, **kw):
            seen["backend_kwargs"] = kw
            return Cap()

        monkeypatch.setattr(scheduler, "get_backend", capture_backend)
        out = scheduler.call_models_parallel(["stub/x"], "SPEC", 1, "tech")
        assert out[0].agreed
        assert seen["backend_kwargs"]["codex_search"] is False
        assert seen["backend_kwargs"]["bedrock_mode"] is False
        assert "previously signalled agreement" not in seen["user"]  # press off
        assert "ERRORS" not in seen["user"]  # preserve-intent off

    def test_build_user_message_defaults(self):
        from adversarial_spec_amd.engine.scheduler import build_user_message

        msg = build_user_message("SPEC", 1, "tech")
        assert "previously signalled agreement" not in msg  # press off
        assert "ERRORS" not in msg  # preserve-intent taxonomy off
