"""Test fixtures: marker registration + filesystem/env isolation.

Mirrors the reference's test strategy: every config-touching
test patches the module-level path constants into a tmpdir; env isolation
via monkeypatch; GPU tests carry @pytest.mark.gpu and only run on MI355X.
"""

from __future__ import annotations

import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X GPU (run via gpurun)")


@pytest.fixture
def isolated_paths(tmp_path, monkeypatch):
    """Redirect sessions/checkpoints/profiles/global-config into tmp_path."""
    from adversarial_spec_amd import providers, session

    sessions = tmp_path / "sessions"
    checkpoints = tmp_path / "checkpoints"
    profiles = tmp_path / "profiles"
    config = tmp_path / "config.json"
    monkeypatch.setattr(session, "SESSIONS_DIR", sessions)
    monkeypatch.setattr(session, "CHECKPOINTS_DIR", checkpoints)
    monkeypatch.setattr(providers, "PROFILES_DIR", profiles)
    monkeypatch.setattr(providers, "GLOBAL_CONFIG_PATH", config)
    return tmp_path


@pytest.fixture
def clean_env(monkeypatch):
    """Strip provider keys / backend overrides from the environment."""
    for var in [
        "OPENAI_API_KEY", "ANTHROPIC_API_KEY", "GEMINI_API_KEY", "XAI_API_KEY",
        "MISTRAL_API_KEY", "GROQ_API_KEY", "OPENROUTER_API_KEY",
        "DEEPSEEK_API_KEY", "ZHIPUAI_API_KEY", "ADVSPEC_BACKEND",
        "ADVSPEC_FORCE_LOCAL", "TELEGRAM_BOT_TOKEN", "TELEGRAM_CHAT_ID",
    ]:
        monkeypatch.delenv(var, raising=False)
    return monkeypatch


@pytest.fixture
def fresh_cost_tracker():
    from adversarial_spec_amd.protocol import cost_tracker

    cost_tracker.reset()
    yield cost_tracker
    cost_tracker.reset()
