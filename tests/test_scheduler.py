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
