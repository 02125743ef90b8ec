"""Backend unit tests with faked subprocess / litellm transports.

Mirrors the reference's fake-backend strategy: canned JSONL
stdout for codex (reference tests: test_models.py:528-671), noise-filtered
plain stdout for gemini-cli (test_models.py:674-789), and routing rules.
"""

from __future__ import annotations

import subprocess
from types import SimpleNamespace
from unittest.mock import patch

import pytest

from adversarial_spec_amd.engine import backend as be
from adversarial_spec_amd.engine.backend import StubBackend


def _run_result(stdout="", stderr="", returncode=0):
    return SimpleNamespace(stdout=stdout, stderr=stderr, returncode=returncode)


class TestCodexBackend:
    CANNED = "\n".join([
        '{"type": "item.started"}',
        "not json at all",
        '{"type": "item.completed", "item": {"type": "other", "text": "nope"}}',
        '{"type": "item.completed", "item": {"type": "agent_message", '
        '"text": "The spec lacks error budgets. [SPEC]better[/SPEC]"}}',
        '{"type": "turn.completed", "usage": {"input_tokens": 321, '
        '"output_tokens": 45}}',
    ])

    def test_parses_jsonl_events(self):
        with patch.object(be, "CODEX_AVAILABLE", True), \
             patch.object(be.subprocess, "run",
                          return_value=_run_result(self.CANNED)) as run:
            text, itok, otok = be.CodexCLIBackend("codex/gpt-5").generate(
                "sys", "user", max_tokens=100, temperature=0.7, timeout=60)
        assert "[SPEC]better[/SPEC]" in text
        assert (itok, otok) == (321, 45)
        argv = run.call_args[0][0]
        assert argv[:3] == ["codex", "exec", "--json"]
        assert "--model" in argv and "gpt-5" in argv

    def test_reasoning_and_search_flags(self):
        with patch.object(be, "CODEX_AVAILABLE", True), \
             patch.object(be.subprocess, "run",
                          return_value=_run_result(self.CANNED)) as run:
            be.CodexCLIBackend("codex/gpt-5", reasoning="low",
                               search=True).generate(
                "s", "u", max_tokens=1, temperature=0, timeout=5)
        argv = run.call_args[0][0]
        assert 'model_reasoning_effort="low"' in argv
        assert "--search" in argv

    def test_no_agent_message_raises(self):
        with patch.object(be, "CODEX_AVAILABLE", True), \
             patch.object(be.subprocess, "run",
                          return_value=_run_result('{"type": "noop"}')):
            with pytest.raises(RuntimeError, match="No agent message"):
                be.CodexCLIBackend("codex/gpt-5").generate(
                    "s", "u", max_tokens=1, temperature=0, timeout=5)

    def test_nonzero_exit_raises_with_stderr(self):
        with patch.object(be, "CODEX_AVAILABLE", True), \
             patch.object(be.subprocess, "run",
                          return_value=_run_result("", "boom", 2)):
            with pytest.raises(RuntimeError, match="boom"):
                be.CodexCLIBackend("codex/gpt-5").generate(
                    "s", "u", max_tokens=1, temperature=0, timeout=5)

    def test_timeout_raises(self):
        with patch.object(be, "CODEX_AVAILABLE", True), \
             patch.object(be.subprocess, "run",
                          side_effect=subprocess.TimeoutExpired("codex", 5)):
            with pytest.raises(RuntimeError, match="timed out"):
                be.CodexCLIBackend("codex/gpt-5").generate(
                    "s", "u", max_tokens=1, temperature=0, timeout=5)

    def test_unavailable_raises(self):
        with patch.object(be, "CODEX_AVAILABLE", False):
            with pytest.raises(RuntimeError, match="not found"):
                be.CodexCLIBackend("codex/gpt-5").generate(
                    "s", "u", max_tokens=1, temperature=0, timeout=5)


class TestGeminiCLIBackend:
    def test_noise_lines_filtered_and_token_estimate(self):
        stdout = "\n".join([
            "Loaded cached credentials.",
            "Server started on port 1234",
            "Loading extension: foo",
            "A real critique line.",
            "[AGREE]",
        ])
        with patch.object(be, "GEMINI_CLI_AVAILABLE", True), \
             patch.object(be.subprocess, "run",
                          return_value=_run_result(stdout)) as run:
            text, itok, otok = be.GeminiCLIBackend("gemini-cli/gemini-2.5-pro").generate(
                "sys", "user msg", max_tokens=10, temperature=0.7, timeout=30)
        assert text == "A real critique line.\n[AGREE]"
        # 4 chars/token estimate (reference: models.py:444-447)
        prompt = be._combined_prompt("sys", "user msg")
        assert itok == len(prompt) // 4
        assert otok == len(text) // 4
        argv = run.call_args[0][0]
        assert argv == ["gemini", "-m", "gemini-2.5-pro", "-y"]
        assert run.call_args[1]["input"] == prompt

    def test_empty_after_filter_raises(self):
        with patch.object(be, "GEMINI_CLI_AVAILABLE", True), \
             patch.object(be.subprocess, "run",
                          return_value=_run_result("Loaded cached creds\n")):
            with pytest.raises(RuntimeError, match="No response"):
                be.GeminiCLIBackend("gemini-cli/x").generate(
                    "s", "u", max_tokens=1, temperature=0, timeout=5)


class TestRouting:
    def test_prefix_routing(self, monkeypatch):
        monkeypatch.delenv("ADVSPEC_BACKEND", raising=False)
        assert isinstance(be.get_backend("stub/critique"), be.StubBackend)
        assert isinstance(be.get_backend("codex/gpt-5"), be.CodexCLIBackend)
        assert isinstance(be.get_backend("gemini-cli/g"), be.GeminiCLIBackend)
        assert isinstance(be.get_backend("gpt-4o"), be.LiteLLMBackend)
        assert isinstance(be.get_backend("local/llama-3-8b"), be.LocalBackend)

    def test_env_forces_stub(self, monkeypatch):
        monkeypatch.setenv("ADVSPEC_BACKEND", "stub")
        assert isinstance(be.get_backend("gpt-4o"), be.StubBackend)

    def test_o_series_detection(self):
        # reference: models.py:50-64 (o1/o3/o4 prefixes, case-insensitive)
        assert be.is_o_series_model("o1-preview")
        assert be.is_o_series_model("openai/o1")
        assert be.is_o_series_model("azure-o1")
        assert not be.is_o_series_model("gpt-4o")


class TestStubContract:
    """Pin the stub backend's simulation contract exactly: other suites
    (retry, fault isolation, CLI flows) depend on these behaviors, so a
    drifting stub silently weakens THEIR assertions (mutation-driven)."""

    def setup_method(self):
        StubBackend.reset_flaky()

    def test_mode_parsed_from_model_string(self):
        assert StubBackend("stub/agree").mode == "agree"
        assert StubBackend("bare").mode == "auto"

    def test_flaky_fails_exactly_twice(self):
        b = StubBackend("stub/flaky")
        kw = dict(max_tokens=10, temperature=0.0, timeout=5.0)
        with pytest.raises(RuntimeError, match="transient failure 1"):
            b.generate("s", "round 1", **kw)
        with pytest.raises(RuntimeError, match="transient failure 2"):
            b.generate("s", "round 1", **kw)
        text, _, _ = b.generate("s", "round 1", **kw)  # third call succeeds
        assert "[SPEC]" in text

    def test_auto_mode_round_threshold(self):
        b = StubBackend("stub/auto")
        kw = dict(max_tokens=10, temperature=0.0, timeout=5.0)
        t1, _, _ = b.generate("s", "This is round 1 of spec dev", **kw)
        assert "[AGREE]" not in t1
        t2, _, _ = b.generate("s", "This is round 2 of spec dev", **kw)
        assert "[AGREE]" in t2

    def test_critique_mode_never_agrees(self):
        b = StubBackend("stub/critique")
        kw = dict(max_tokens=10, temperature=0.0, timeout=5.0)
        t, _, _ = b.generate("s", "This is round 9 of spec dev", **kw)
        assert "[AGREE]" not in t and "[SPEC]" in t

    def test_token_accounting_is_chars_div_4(self):
        b = StubBackend("stub/agree")
        text, in_tok, out_tok = b.generate(
            "x" * 40, "This is round 1" + "y" * 25,
            max_tokens=10, temperature=0.0, timeout=5.0)
        assert in_tok == (40 + len("This is round 1" + "y" * 25)) // 4
        assert out_tok == len(text) // 4

    def test_round_parse_beyond_header_window(self):
        # round number read from the first 64 chars only (header window)
        rnd = StubBackend._round_from_message("This is round 7 of x")
        assert rnd == 7
        assert StubBackend._round_from_message("no digits here") == 1


class TestCLIBackendCmdConstruction:
    """The subprocess backends must pass the BARE model name (prefix
    stripped) to the CLI binary."""

    def test_codex_model_name_stripped(self, monkeypatch):
        from adversarial_spec_amd.engine import backend as B

        monkeypatch.setattr(B, "CODEX_AVAILABLE", True)
        seen = {}

        def fake_run(cmd, **kw):
            seen["cmd"] = cmd
            class R:
                returncode = 0
                stdout = ('{"type": "item.completed", "item": '
                          '{"type": "agent_message", "text": "ok"}}')
                stderr = ""
            return R()

        monkeypatch.setattr(B.subprocess, "run", fake_run)
        b = B.CodexCLIBackend("codex/gpt-5.1-codex")
        text, _, _ = b.generate("s", "u", max_tokens=1, temperature=0.0,
                                timeout=5.0)
        assert text == "ok"
        i = seen["cmd"].index("--model")
        assert seen["cmd"][i + 1] == "gpt-5.1-codex"  # prefix stripped

    def test_gemini_model_name_stripped(self, monkeypatch):
        from adversarial_spec_amd.engine import backend as B

        monkeypatch.setattr(B, "GEMINI_CLI_AVAILABLE", True)
        seen = {}

        def fake_run(cmd, **kw):
            seen["cmd"] = cmd
            class R:
                returncode = 0
                stdout = "fine"
                stderr = ""
            return R()

        monkeypatch.setattr(B.subprocess, "run", fake_run)
        b = B.GeminiCLIBackend("gemini-cli/gemini-3-pro")
        b.generate("s", "u", max_tokens=1, temperature=0.0, timeout=5.0)
        i = seen["cmd"].index("-m")
        assert seen["cmd"][i + 1] == "gemini-3-pro"


class TestBackendDefaultsAndRouting:
    """Mutation killers: flag defaults and maxsplit=1 model-name routing."""

    def test_defaults_are_off(self):
        # opt-in flags must DEFAULT off: search hits the network, bedrock
        # rewrites model ids
        assert be.CodexCLIBackend("codex/m").search is False
        assert be.LiteLLMBackend("m").bedrock_mode is False
        cb = be.get_backend("codex/m")
        assert cb.search is False
        lb = be.get_backend("claude-sonnet-4")
        assert lb.bedrock_mode is False

    def test_multislash_names_keep_tail(self):
        # split("/", 1): everything after the FIRST slash is the name
        # (gemini model paths legitimately contain slashes)
        assert StubBackend("stub/a/b").mode == "a/b"
        with pytest.raises(ValueError, match="a/b"):
            be.LocalBackend("local/a/b")

    def test_codex_subprocess_kwargs(self, monkeypatch):
        monkeypatch.setattr(be, "CODEX_AVAILABLE", True)
        seen = {}

        def fake_run(cmd, **kw):
            seen.update(kw, cmd=cmd)
            class R:
                returncode = 0
                stdout = ('{"type": "item.completed", "item": '
                          '{"type": "agent_message", "text": "ok"}}')
                stderr = ""
            return R()

        monkeypatch.setattr(be.subprocess, "run", fake_run)
        be.CodexCLIBackend("codex/a/b").generate(
            "s", "u", max_tokens=1, temperature=0.0, timeout=5.0)
        i = seen["cmd"].index("--model")
        assert seen["cmd"][i + 1] == "a/b"  # maxsplit=1 tail
        assert seen["capture_output"] is True and seen["text"] is True

    def test_gemini_subprocess_kwargs(self, monkeypatch):
        monkeypatch.setattr(be, "GEMINI_CLI_AVAILABLE", True)
        seen = {}

        def fake_run(cmd, **kw):
            seen.update(kw, cmd=cmd)
            class R:
                returncode = 0
                stdout = "fine"
                stderr = ""
            return R()

        monkeypatch.setattr(be.subprocess, "run", fake_run)
        be.GeminiCLIBackend("gemini-cli/models/gx").generate(
            "s", "u", max_tokens=1, temperature=0.0, timeout=5.0)
        i = seen["cmd"].index("-m")
        assert seen["cmd"][i + 1] == "models/gx"
        assert seen["capture_output"] is True and seen["text"] is True
