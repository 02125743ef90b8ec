"""CLI-level tests driving main() with patched argv/stdin."""

import io
import json
from unittest.mock import patch

import pytest

from adversarial_spec_amd.cli import debate
from adversarial_spec_amd.protocol import ModelResponse


def run_cli(argv, stdin=""):
    out, err = io.StringIO(), io.StringIO()
    with patch("sys.stdin", io.StringIO(stdin)), \
         patch("sys.stdout", out), patch("sys.stderr", err):
        try:
            code = debate.main(argv)
        except SystemExit as e:
            code = e.code
    return code, out.getvalue(), err.getvalue()


class TestInfoActions:
    def test_providers(self, isolated_paths, clean_env):
        code, out, _ = run_cli(["providers"])
        assert code == 0
        assert "Local (MI355X) model registry" in out

    def test_focus_areas(self):
        code, out, _ = run_cli(["focus-areas"])
        assert code == 0
        for f in ("security", "scalability", "performance", "ux", "reliability", "cost"):
            assert f in out

    def test_personas(self):
        code, out, _ = run_cli(["personas"])
        assert code == 0
        assert "security-engineer" in out

    def test_profiles_empty(self, isolated_paths):
        code, out, _ = run_cli(["profiles"])
        assert code == 0
        assert "No saved profiles" in out

    def test_sessions_empty(self, isolated_paths):
        code, out, _ = run_cli(["sessions"])
        assert code == 0
        assert "No saved sessions" in out


class TestUtilityActions:
    def test_diff(self, tmp_path):
        a = tmp_path / "a.md"
        b = tmp_path / "b.md"
        a.write_text("one\ntwo\n")
        b.write_text("one\nthree\n")
        code, out, _ = run_cli(["diff", "--previous", str(a), "--current", str(b)])
        assert code == 0
        assert "-two" in out and "+three" in out

    def test_diff_missing_flags(self):
        code, _, err = run_cli(["diff"])
        assert code == 1

    def test_save_profile_and_list(self, isolated_paths):
        code, out, _ = run_cli(
            ["save-profile", "mine", "--models", "stub/agree", "--focus", "security"]
        )
        assert code == 0
        code, out, _ = run_cli(["profiles"])
        assert "mine" in out

    def test_bedrock_status(self, isolated_paths):
        code, out, _ = run_cli(["bedrock", "status"])
        assert code == 0
        assert "disabled" in out

    def test_local_status(self, isolated_paths, clean_env):
        code, out, _ = run_cli(["local", "status"])
        assert code == 0

    def test_local_alias_flow(self, isolated_paths, clean_env):
        code, out, _ = run_cli(
            ["local", "alias", "mine", "llama-3-8b", "--weights", "/w", "--gpu", "1"]
        )
        assert code == 0
        from adversarial_spec_amd.providers import get_local_config

        assert get_local_config()["custom_aliases"]["mine"]["gpu"] == 1


class TestCritique:
    def test_no_stdin_exits_1(self, isolated_paths, clean_env):
        code, _, err = run_cli(["critique", "--models", "stub/agree"], stdin="")
        assert code == 1
        assert "No spec provided" in err

    def test_missing_credentials_exit_2(self, isolated_paths, clean_env):
        import torch

        if torch.cuda.is_available():
            pytest.skip("GPU present: local engine is a valid backend")
        code, _, err = run_cli(["critique", "--models", "gpt-4o"], stdin="spec")
        assert code == 2
        assert "missing credentials" in err.lower() or "Error" in err

    def test_json_schema(self, isolated_paths, clean_env, fresh_cost_tracker):
        code, out, err = run_cli(
            ["critique", "--models", "stub/critique,stub/agree", "--json"],
            stdin="# spec body",
        )
        assert code == 0
        data = json.loads(out)
        assert set(data) >= {
            "all_agreed", "round", "doc_type", "models", "focus", "persona",
            "preserve_intent", "session", "results", "cost",
        }
        assert data["all_agreed"] is False
        assert data["models"] == ["stub/critique", "stub/agree"]
        r0 = data["results"][0]
        assert set(r0) == {
            "model", "agreed", "response", "spec", "error",
            "input_tokens", "output_tokens", "cost",
        }
        assert data["cost"]["total"] >= 0

    def test_all_agree_text_output(self, isolated_paths, clean_env, fresh_cost_tracker):
        code, out, _ = run_cli(
            ["critique", "--models", "stub/agree,stub/agree"], stdin="spec"
        )
        assert code == 0
        assert "ALL MODELS AGREE" in out

    def test_partial_agreement_lists(self, isolated_paths, clean_env, fresh_cost_tracker):
        code, out, _ = run_cli(
            ["critique", "--models", "stub/agree,stub/critique"], stdin="spec"
        )
        assert "Agreed: stub/agree" in out
        assert "Critiqued: stub/critique" in out

    def test_errors_excluded_from_consensus(self, isolated_paths, clean_env,
                                            fresh_cost_tracker):
        with patch("adversarial_spec_amd.engine.scheduler.time.sleep"):
            code, out, err = run_cli(
                ["critique", "--models", "stub/agree,stub/error", "--json"],
                stdin="spec",
            )
        data = json.loads(out)
        assert data["all_agreed"] is True  # errors excluded (reference debate.py:853)
        assert data["results"][1]["error"]
        assert "returned error" in err

    def test_option_plumb_through(self, isolated_paths, clean_env, fresh_cost_tracker):
        captured = {}

        def fake_parallel(models, spec, round_num, doc_type, press, focus, persona,
                          context, preserve_intent, codex_reasoning, codex_search,
                          timeout, bedrock_mode, bedrock_region):
            captured.update(
                models=models, round=round_num, doc_type=doc_type, press=press,
                focus=focus, persona=persona, preserve_intent=preserve_intent,
                timeout=timeout, codex_reasoning=codex_reasoning,
            )
            return [ModelResponse(model=m, response="[AGREE]", agreed=True)
                    for m in models]

        with patch.object(debate, "call_models_parallel", fake_parallel):
            code, out, _ = run_cli(
                ["critique", "--models", "stub/x", "--round", "4", "--doc-type", "prd",
                 "--press", "--focus", "security", "--persona", "qa-engineer",
                 "--preserve-intent", "--timeout", "33",
                 "--codex-reasoning", "low", "--json"],
                stdin="spec",
            )
        assert code == 0
        assert captured["round"] == 4
        assert captured["doc_type"] == "prd"
        assert captured["press"] is True
        assert captured["focus"] == "security"
        assert captured["persona"] == "qa-engineer"
        assert captured["preserve_intent"] is True
        assert captured["timeout"] == 33
        assert captured["codex_reasoning"] == "low"

    def test_show_cost(self, isolated_paths, clean_env, fresh_cost_tracker):
        code, out, _ = run_cli(
            ["critique", "--models", "stub/agree", "--show-cost"], stdin="spec"
        )
        assert "Cost Summary" in out


class TestSessionFlow:
    def test_session_created_and_resumed(self, isolated_paths, clean_env,
                                         fresh_cost_tracker, monkeypatch, tmp_path):
        import os

        monkeypatch.chdir(tmp_path)
        code, out, err = run_cli(
            ["critique", "--models", "stub/critique", "--session", "s1", "--json"],
            stdin="original spec",
        )
        assert code == 0
        assert "Session 's1' created" in err
        # checkpoint written (CHECKPOINTS_DIR redirected by isolated_paths)
        assert (isolated_paths / "checkpoints" / "s1-round-1.md").exists()
        from adversarial_spec_amd.session import SessionState

        st = SessionState.load("s1")
        assert st.round == 2  # advanced
        assert st.history[0]["round"] == 1
        assert st.spec != "original spec"  # revised spec persisted

        # resume (no stdin needed)
        code2, out2, err2 = run_cli(["critique", "--resume", "s1", "--json"], stdin="")
        assert code2 == 0
        assert "Resuming session 's1' at round 2" in err2
        data = json.loads(out2)
        assert data["round"] == 2

    def test_resume_missing_exit_2(self, isolated_paths, clean_env):
        code, _, err = run_cli(["critique", "--resume", "ghost"], stdin="")
        assert code == 2


class TestExportTasks:
    def test_export_tasks_json(self, isolated_paths, clean_env, fresh_cost_tracker):
        canned = (
            "[TASK]\ntitle: Do thing\ntype: task\npriority: high\n"
            "description: details\nacceptance_criteria:\n- done\n[/TASK]"
        )

        class FakeBackend:
            def generate(self, *a, **k):
                return canned, 10, 10

        with patch("adversarial_spec_amd.engine.backend.get_backend",
                   return_value=FakeBackend()):
            code, out, _ = run_cli(
                ["export-tasks", "--models", "stub/x", "--json"], stdin="spec"
            )
        assert code == 0
        data = json.loads(out)
        assert data["tasks"][0]["title"] == "Do thing"

    def test_export_tasks_no_stdin(self, isolated_paths, clean_env):
        code, _, err = run_cli(["export-tasks", "--models", "stub/x"], stdin="")
        assert code == 1


class TestTelegramGlue:
    def test_send_final_unconfigured_exit_2(self, isolated_paths, clean_env):
        code, _, err = run_cli(
            ["send-final", "--models", "stub/x", "--rounds", "3"], stdin="final spec"
        )
        assert code == 2

    def test_critique_with_telegram_feedback(self, isolated_paths, clean_env,
                                             fresh_cost_tracker):
        clean_env.setenv("TELEGRAM_BOT_TOKEN", "t")
        clean_env.setenv("TELEGRAM_CHAT_ID", "42")
        with patch("adversarial_spec_amd.telegram.get_last_update_id", return_value=5), \
             patch("adversarial_spec_amd.telegram.send_long_message", return_value=True), \
             patch("adversarial_spec_amd.telegram.poll_for_reply",
                   return_value="add a rollback section"):
            code, out, _ = run_cli(
                ["critique", "--models", "stub/agree", "--telegram", "--json"],
                stdin="spec",
            )
        data = json.loads(out)
        assert data["user_feedback"] == "add a rollback section"


class TestMutationKillers:
    """Pin precedence/consensus/dispatch exactness (mutation-driven)."""

    def test_profile_doc_type_precedence(self, clean_env, tmp_path, monkeypatch):
        """A profile's doc_type fills only the DEFAULT (tech); an explicit
        --doc-type wins over the profile."""
        from adversarial_spec_amd import providers

        monkeypatch.setattr(providers, "PROFILES_DIR", tmp_path)
        providers.save_profile("p", {"doc_type": "prd"})

        parser = debate.create_parser()
        args = parser.parse_args(["critique", "--profile", "p"])
        debate.apply_profile(args)
        assert args.doc_type == "prd"  # default tech -> profile fills

        args = parser.parse_args(
            ["critique", "--profile", "p", "--doc-type", "tech"])
        # explicit tech is indistinguishable from the default by design
        # (reference semantics, debate.py:529-550) — but an explicit PRD
        # must never be overridden the other way:
        providers.save_profile("p2", {"doc_type": "tech"})
        args = parser.parse_args(
            ["critique", "--profile", "p2", "--doc-type", "prd"])
        debate.apply_profile(args)
        assert args.doc_type == "prd"

    def test_all_error_round_is_not_consensus(self, clean_env):
        """Every opponent erroring must NOT report all_agreed (a round with
        zero successful critiques has no consensus to claim)."""
        code, out, err = run_cli(
            ["critique", "--models", "stub/error", "--json"], stdin="spec")
        data = json.loads(out)
        assert data["all_agreed"] is False

    def test_round_default_is_1(self):
        args = debate.create_parser().parse_args(["critique"])
        assert args.round == 1

    def test_export_tasks_numbering_starts_at_1(self, clean_env):
        canned = ("[TASK]\ntitle: First\ntype: feature\npriority: high\n"
                  "description: d\n[/TASK]")

        class FakeBackend:
            def generate(self, *a, **k):
                return canned, 10, 10

        with patch("adversarial_spec_amd.engine.backend.get_backend",
                   return_value=FakeBackend()):
            code, out, _ = run_cli(
                ["export-tasks", "--models", "stub/x"], stdin="spec")
        assert code == 0
        assert "1. [feature] [high] First" in out

    def test_daemon_forward_uses_sys_argv_tail(self, clean_env, monkeypatch):
        """main(None) forwards sys.argv[1:] EXACTLY to a live daemon ([2:]
        would eat the action and replay the wrong command)."""
        import sys as _sys

        from adversarial_spec_amd import daemon as dmod

        argv = ["debate.py", "critique", "--models", "stub/agree", "--json"]
        monkeypatch.setattr(_sys, "argv", argv)
        seen = {}

        def fake_forward(raw_argv, stdin_text, socket_path=None):
            seen["argv"] = list(raw_argv)
            return 0, "{}", ""

        monkeypatch.setattr(dmod, "ping", lambda *a, **k: True)
        monkeypatch.setattr(dmod, "try_forward", fake_forward)
        with patch("sys.stdin", io.StringIO("spec")), \
                patch("sys.stdout", io.StringIO()):
            code = debate.main(None)
        assert code == 0
        assert seen["argv"] == argv[1:]

    def test_setup_bedrock_enabled_returns_true(self, isolated_paths, clean_env):
        """With bedrock enabled and models valid, setup_bedrock must report
        bedrock_mode=True (False would silently route to plain litellm)."""
        from adversarial_spec_amd import providers

        providers.handle_bedrock_command("enable", None, None, "us-west-2")
        providers.handle_bedrock_command("add-model", "llama-3-8b", None, None)
        args = debate.create_parser().parse_args(["critique"])
        mode, region = debate.setup_bedrock(args, ["llama-3-8b"])
        assert mode is True
        assert region == "us-west-2"

    def test_send_final_rounds_default_is_1(self):
        args = debate.create_parser().parse_args(["send-final"])
        assert args.rounds == 1
