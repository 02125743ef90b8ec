"""Provider/config/registry tests with filesystem + env isolation."""

import json

from adversarial_spec_amd import providers


class TestCosts:
    def test_known_model(self):
        assert providers.get_model_cost("gpt-4o")["input"] == 2.50

    def test_unknown_default(self):
        assert providers.get_model_cost("nope") == providers.DEFAULT_COST

    def test_local_free(self):
        assert providers.get_model_cost("local/llama-3-8b") == {"input": 0.0, "output": 0.0}


class TestGlobalConfig:
    def test_roundtrip(self, isolated_paths):
        providers.save_global_config({"bedrock": {"enabled": True}})
        assert providers.load_global_config()["bedrock"]["enabled"] is True

    def test_missing_is_empty(self, isolated_paths):
        assert providers.load_global_config() == {}

    def test_corrupt_is_empty(self, isolated_paths):
        providers.GLOBAL_CONFIG_PATH.parent.mkdir(parents=True, exist_ok=True)
        providers.GLOBAL_CONFIG_PATH.write_text("{not json")
        assert providers.load_global_config() == {}


class TestBedrock:
    def test_resolve_map(self):
        assert providers.resolve_bedrock_model("llama-3-8b", {}) == "meta.llama3-8b-instruct-v1:0"

    def test_resolve_alias_wins(self):
        cfg = {"custom_aliases": {"llama-3-8b": "custom.id"}}
        assert providers.resolve_bedrock_model("llama-3-8b", cfg) == "custom.id"

    def test_resolve_passthrough(self):
        assert providers.resolve_bedrock_model("raw.model.id", {}) == "raw.model.id"

    def test_validate(self):
        cfg = {"available_models": ["claude-3-sonnet"], "custom_aliases": {"my": "x"}}
        valid, invalid = providers.validate_bedrock_models(
            ["claude-3-sonnet", "bedrock/my", "nope"], cfg
        )
        assert valid == ["claude-3-sonnet", "bedrock/my"]
        assert invalid == ["nope"]

    def test_subcommands(self, isolated_paths, capsys):
        assert providers.handle_bedrock_command("enable", None, None, "us-east-1") == 0
        assert providers.get_bedrock_config()["enabled"] is True
        assert providers.handle_bedrock_command("add-model", "claude-3-haiku", None, None) == 0
        assert "claude-3-haiku" in providers.get_bedrock_config()["available_models"]
        assert providers.handle_bedrock_command("alias", "my", "vendor.id", None) == 0
        assert providers.get_bedrock_config()["custom_aliases"]["my"] == "vendor.id"
        assert providers.handle_bedrock_command("remove-model", "claude-3-haiku", None, None) == 0
        assert providers.handle_bedrock_command("disable", None, None, None) == 0
        assert providers.get_bedrock_config()["enabled"] is False
        capsys.readouterr()

    def test_enable_without_region_fails(self, isolated_paths, capsys):
        assert providers.handle_bedrock_command("enable", None, None, None) == 1
        capsys.readouterr()


class TestLocalRegistry:
    def test_resolve_known(self):
        spec = providers.resolve_local_model("llama-3-8b", {})
        assert spec["arch"] == "llama-3-8b" and spec["weights"] is None

    def test_resolve_alias(self):
        cfg = {"custom_aliases": {"mine": {"arch": "llama-3-8b", "weights": "/w", "gpu": 3}}}
        spec = providers.resolve_local_model("mine", cfg)
        assert spec["weights"] == "/w" and spec["gpu"] == 3

    def test_resolve_unknown_raises(self):
        import pytest

        with pytest.raises(ValueError):
            providers.resolve_local_model("doesnotexist", {})

    def test_subcommands(self, isolated_paths, capsys):
        assert providers.handle_local_command("add-model", "llama-3-8b", None, None, None) == 0
        assert "llama-3-8b" in providers.get_local_config()["available_models"]
        assert providers.handle_local_command(
            "alias", "mine", "llama-3-8b", "/weights/dir", 2
        ) == 0
        alias = providers.get_local_config()["custom_aliases"]["mine"]
        assert alias == {"arch": "llama-3-8b", "weights": "/weights/dir", "gpu": 2}
        assert providers.handle_local_command("remove-model", "llama-3-8b", None, None, None) == 0
        capsys.readouterr()

    def test_add_unknown_fails(self, isolated_paths, capsys):
        assert providers.handle_local_command("add-model", "nope", None, None, None) == 1
        capsys.readouterr()


class TestProfiles:
    def test_roundtrip(self, isolated_paths):
        providers.save_profile("p1", {"models": "local/llama-3-8b", "focus": "security"})
        data = providers.load_profile("p1")
        assert data["models"] == "local/llama-3-8b"
        assert data["focus"] == "security"
        assert "p1" in providers.list_profiles()

    def test_missing_none(self, isolated_paths):
        assert providers.load_profile("nope") is None

    def test_traversal_guard(self, isolated_paths):
        assert providers.load_profile("../../etc/passwd") is None


class TestDiscovery:
    def test_no_keys_no_gpu(self, clean_env):
        import torch

        if torch.cuda.is_available():
            return  # covered by gpu-marked variants
        avail = providers.get_available_providers()
        names = [a[0] for a in avail]
        assert "OpenAI" not in names

    def test_key_detected(self, clean_env):
        clean_env.setenv("OPENAI_API_KEY", "sk-xxx")
        names = [a[0] for a in providers.get_available_providers()]
        assert "OpenAI" in names

    def test_default_model_priority(self, clean_env, isolated_paths):
        clean_env.setenv("ADVSPEC_FORCE_LOCAL", "1")
        assert providers.get_default_model() == "local/llama-3-8b"

    def test_default_model_from_key(self, clean_env, isolated_paths):
        import torch

        if torch.cuda.is_available():
            return
        clean_env.setenv("GEMINI_API_KEY", "k")
        assert providers.get_default_model() == "gemini/gemini-2.0-flash"


class TestValidateCredentials:
    def test_local_with_force(self, clean_env, isolated_paths):
        clean_env.setenv("ADVSPEC_FORCE_LOCAL", "1")
        valid, invalid = providers.validate_model_credentials(["local/llama-3-8b"])
        assert valid == ["local/llama-3-8b"]

    def test_local_unknown_invalid(self, clean_env, isolated_paths):
        clean_env.setenv("ADVSPEC_FORCE_LOCAL", "1")
        valid, invalid = providers.validate_model_credentials(["local/bogus"])
        assert invalid == ["local/bogus"]

    def test_api_key_missing(self, clean_env, isolated_paths):
        valid, invalid = providers.validate_model_credentials(["gpt-4o"])
        assert invalid == ["gpt-4o"]

    def test_api_key_present(self, clean_env, isolated_paths):
        clean_env.setenv("OPENAI_API_KEY", "sk")
        valid, invalid = providers.validate_model_credentials(["gpt-4o", "o1-mini"])
        assert valid == ["gpt-4o", "o1-mini"]

    def test_unknown_scheme_passes(self, clean_env, isolated_paths):
        valid, invalid = providers.validate_model_credentials(["weird/model"])
        assert valid == ["weird/model"]


class TestMutationKillers:
    """Pin exact exit codes / defaults the example tests left unobserved."""

    def test_bedrock_error_paths_exit_1_exactly(self, isolated_paths, capsys):
        h = providers.handle_bedrock_command
        assert h("add-model", None, None, None) == 1
        assert h("remove-model", None, None, None) == 1
        assert h("remove-model", "not-there", None, None) == 1
        assert h("alias", "name-only", None, None) == 1
        assert h("definitely-unknown", None, None, None) == 1
        capsys.readouterr()

    def test_bedrock_list_models_exits_0(self, isolated_paths, capsys):
        assert providers.handle_bedrock_command("list-models", None, None,
                                                None) == 0
        out = capsys.readouterr().out
        assert "llama-3-8b" in out

    def test_local_error_paths_exit_1_exactly(self, isolated_paths, capsys):
        h = providers.handle_local_command
        assert h("add-model", None, None, None, None) == 1
        assert h("add-model", "no-such-arch-xyz", None, None, None) == 1
        assert h("remove-model", None, None, None, None) == 1
        assert h("remove-model", "not-there", None, None, None) == 1
        assert h("alias", "name-only", None, None, None) == 1
        assert h("alias", "my", "no-such-arch-xyz", None, None) == 1
        assert h("definitely-unknown", None, None, None, None) == 1
        capsys.readouterr()

    def test_local_list_models_exits_0(self, isolated_paths, capsys):
        assert providers.handle_local_command("list-models", None, None,
                                              None, None) == 0
        assert "llama-3-8b" in capsys.readouterr().out

    def test_bedrock_defaults_disabled(self, isolated_paths, capsys):
        """A fresh config materialized by any bedrock write must default to
        enabled: False (opt-in)."""
        assert providers.handle_bedrock_command("add-model", "llama-3-8b",
                                                None, None) == 0
        cfg = providers.load_global_config()
        assert cfg["bedrock"]["enabled"] is False
        capsys.readouterr()

    def test_config_saves_survive_existing_and_deep_dirs(
            self, isolated_paths, tmp_path, monkeypatch):
        """save_global_config / save_profile must create parents (a fresh
        install has NO ~/.claude/adversarial-spec chain) AND tolerate the
        directory already existing (second save)."""
        monkeypatch.setattr(providers, "GLOBAL_CONFIG_PATH",
                            tmp_path / "deep" / "chain" / "config.json")
        monkeypatch.setattr(providers, "PROFILES_DIR",
                            tmp_path / "deep" / "profiles" / "nested")
        providers.save_global_config({"a": 1})
        providers.save_global_config({"a": 2})  # dir exists now
        assert providers.load_global_config() == {"a": 2}
        providers.save_profile("p1", {"models": ["x"]})
        providers.save_profile("p1", {"models": ["y"]})
        assert providers.load_profile("p1")["models"] == ["y"]

    def test_local_engine_available_false_on_torch_error(self, monkeypatch):
        monkeypatch.delenv("ADVSPEC_FORCE_LOCAL", raising=False)
        import torch

        def boom():
            raise RuntimeError("no device")

        monkeypatch.setattr(torch.cuda, "is_available", boom)
        assert providers.local_engine_available() is False
