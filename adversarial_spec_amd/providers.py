"""Providers, credentials, cost table, global config, profiles, registries.

Config tiers (reference: SURVEY.md §5.6), later wins:
  1. CLI flags
  2. profiles at ~/.config/adversarial-spec/profiles/<name>.json
  3. global config at ~/.claude/adversarial-spec/config.json

The global config keeps the reference's `bedrock` section shape
(reference: providers.py:88-102, SKILL.md:109-120) and adds a sibling
`local` section for the MI355X on-node model registry, mirroring the
BEDROCK_MODEL_MAP friendly-name semantics (reference: providers.py:57-85):

  {
    "bedrock": {"enabled": bool, "region": str,
                 "available_models": [...], "custom_aliases": {...}},
    "local":   {"enabled": bool,
                 "available_models": ["llama-3-8b", ...],
                 "custom_aliases": {"mymodel": {"arch": "llama-3-8b",
                                                 "weights": "/path", "gpu": 0}}}
  }

Model strings route by prefix (reference: models.py:506/558):
  local/<name>     -> MI355X on-node inference engine (this framework's L1)
  codex/<m>        -> codex CLI subprocess
  gemini-cli/<m>   -> gemini CLI subprocess
  bedrock/<m>      -> litellm bedrock (auto-prefixed in bedrock mode)
  anything else    -> litellm remote API
"""

from __future__ import annotations

import json
import os
import shutil
import sys
from pathlib import Path
from typing import Any, Optional

from .prompts import FOCUS_AREAS, PERSONAS

# ---------------------------------------------------------------------------
# Cost table ($ per 1M tokens). Static pricing facts, same reporting surface
# as the reference (providers.py:18-45). local/ models are billed at 0 like
# the subscription CLIs; GPU-seconds are reported separately in timings.
# ---------------------------------------------------------------------------

MODEL_COSTS: dict[str, dict[str, float]] = {
    "gpt-4o": {"input": 2.50, "output": 10.00},
    "gpt-4-turbo": {"input": 10.00, "output": 30.00},
    "gpt-4": {"input": 30.00, "output": 60.00},
    "gpt-3.5-turbo": {"input": 0.50, "output": 1.50},
    "o1": {"input": 15.00, "output": 60.00},
    "o1-mini": {"input": 3.00, "output": 12.00},
    "claude-sonnet-4-20250514": {"input": 3.00, "output": 15.00},
    "claude-opus-4-20250514": {"input": 15.00, "output": 75.00},
    "gemini/gemini-2.0-flash": {"input": 0.075, "output": 0.30},
    "gemini/gemini-pro": {"input": 0.50, "output": 1.50},
    "xai/grok-3": {"input": 3.00, "output": 15.00},
    "xai/grok-beta": {"input": 5.00, "output": 15.00},
    "mistral/mistral-large": {"input": 2.00, "output": 6.00},
    "groq/llama-3.3-70b-versatile": {"input": 0.59, "output": 0.79},
    "deepseek/deepseek-chat": {"input": 0.14, "output": 0.28},
    "zhipu/glm-4": {"input": 1.40, "output": 1.40},
    "zhipu/glm-4-plus": {"input": 7.00, "output": 7.00},
    # Subscription CLIs: no per-token cost (reference: providers.py:37-42)
    "codex/gpt-5.2-codex": {"input": 0.0, "output": 0.0},
    "codex/gpt-5.1-codex-max": {"input": 0.0, "output": 0.0},
    "codex/gpt-5.1-codex-mini": {"input": 0.0, "output": 0.0},
    "gemini-cli/gemini-3-pro-preview": {"input": 0.0, "output": 0.0},
    "gemini-cli/gemini-3-flash-preview": {"input": 0.0, "output": 0.0},
}

DEFAULT_COST = {"input": 5.00, "output": 15.00}


def get_model_cost(model: str) -> dict[str, float]:
    """$/1M-token rates for a model; local/ models cost 0, unknown = default."""
    if model.startswith("local/"):
        return {"input": 0.0, "output": 0.0}
    return MODEL_COSTS.get(model, DEFAULT_COST)


CODEX_AVAILABLE = shutil.which("codex") is not None
GEMINI_CLI_AVAILABLE = shutil.which("gemini") is not None

DEFAULT_CODEX_REASONING = "xhigh"

# ---------------------------------------------------------------------------
# Bedrock friendly-name map (reference: providers.py:57-85). Kept verbatim in
# semantics: resolve_bedrock_model() checks custom aliases first, then this
# map, then passes through raw IDs.
# ---------------------------------------------------------------------------

BEDROCK_MODEL_MAP: dict[str, str] = {
    "claude-3-sonnet": "anthropic.claude-3-sonnet-20240229-v1:0",
    "claude-3-haiku": "anthropic.claude-3-haiku-20240307-v1:0",
    "claude-3-opus": "anthropic.claude-3-opus-20240229-v1:0",
    "claude-3.5-sonnet": "anthropic.claude-3-5-sonnet-20240620-v1:0",
    "claude-3.5-sonnet-v2": "anthropic.claude-3-5-sonnet-20241022-v2:0",
    "claude-3.5-haiku": "anthropic.claude-3-5-haiku-20241022-v1:0",
    "llama-3-8b": "meta.llama3-8b-instruct-v1:0",
    "llama-3-70b": "meta.llama3-70b-instruct-v1:0",
    "llama-3.1-8b": "meta.llama3-1-8b-instruct-v1:0",
    "llama-3.1-70b": "meta.llama3-1-70b-instruct-v1:0",
    "llama-3.1-405b": "meta.llama3-1-405b-instruct-v1:0",
    "mistral-7b": "mistral.mistral-7b-instruct-v0:2",
    "mistral-large": "mistral.mistral-large-2402-v1:0",
    "mixtral-8x7b": "mistral.mixtral-8x7b-instruct-v0:1",
    "titan-text-express": "amazon.titan-text-express-v1",
    "titan-text-lite": "amazon.titan-text-lite-v1",
    "cohere-command": "cohere.command-text-v14",
    "cohere-command-light": "cohere.command-light-text-v14",
    "cohere-command-r": "cohere.command-r-v1:0",
    "cohere-command-r-plus": "cohere.command-r-plus-v1:0",
    "ai21-jamba": "ai21.jamba-instruct-v1:0",
}

# ---------------------------------------------------------------------------
# Local (MI355X) friendly-name map: name -> architecture preset known to
# adversarial_spec_amd.models.config. Weights default to random-init when no
# path is registered (synthetic/bench mode); a custom alias may pin a
# safetensors directory and a GPU ordinal.
# ---------------------------------------------------------------------------

LOCAL_MODEL_MAP: dict[str, str] = {
    "llama-3-8b": "llama-3-8b",
    "llama-3-70b": "llama-3-70b",
    "llama-3.1-8b": "llama-3-8b",
    "llama-3.1-70b": "llama-3-70b",
    "mistral-7b": "mistral-7b",
    "tiny": "tiny",          # 4-layer CPU test model
    "debug-1b": "debug-1b",  # 16-layer GPU smoke/test model
}

# ---------------------------------------------------------------------------
# Global config (path compat: reference providers.py:15)
# ---------------------------------------------------------------------------

GLOBAL_CONFIG_PATH = Path.home() / ".claude" / "adversarial-spec" / "config.json"


def load_global_config() -> dict[str, Any]:
    if GLOBAL_CONFIG_PATH.exists():
        try:
            return json.loads(GLOBAL_CONFIG_PATH.read_text())
        except (json.JSONDecodeError, OSError):
            return {}
    return {}


def save_global_config(config: dict[str, Any]) -> None:
    GLOBAL_CONFIG_PATH.parent.mkdir(parents=True, exist_ok=True)
    GLOBAL_CONFIG_PATH.write_text(json.dumps(config, indent=2))


def get_bedrock_config() -> dict[str, Any]:
    return load_global_config().get("bedrock", {})


def get_local_config() -> dict[str, Any]:
    return load_global_config().get("local", {})


def resolve_bedrock_model(model: str, config: Optional[dict] = None) -> str:
    """Friendly name -> Bedrock ID: custom aliases, then the static map,
    then pass-through (reference: providers.py:117-145)."""
    if config is None:
        config = get_bedrock_config()
    aliases = config.get("custom_aliases", {})
    if model in aliases:
        return aliases[model]
    if model in BEDROCK_MODEL_MAP:
        return BEDROCK_MODEL_MAP[model]
    return model


def resolve_local_model(name: str, config: Optional[dict] = None) -> dict[str, Any]:
    """Resolve a local/<name> model to an engine spec dict.

    Returns {"name", "arch", "weights", "gpu"}; custom aliases override the
    static map; unknown names raise ValueError.
    """
    if config is None:
        config = get_local_config()
    aliases = config.get("custom_aliases", {})
    if name in aliases:
        a = dict(aliases[name])
        a.setdefault("name", name)
        a.setdefault("arch", LOCAL_MODEL_MAP.get(name, name))
        a.setdefault("weights", None)
        a.setdefault("gpu", None)
        return a
    if name in LOCAL_MODEL_MAP:
        return {"name": name, "arch": LOCAL_MODEL_MAP[name], "weights": None, "gpu": None}
    raise ValueError(
        f"Unknown local model '{name}'. Known: {', '.join(sorted(LOCAL_MODEL_MAP))} "
        f"or register an alias: debate.py local alias {name} <arch> [--weights PATH]"
    )


def validate_bedrock_models(
    models: list[str], config: Optional[dict] = None
) -> tuple[list[str], list[str]]:
    """Split models into (valid, invalid) against the bedrock allow-list."""
    if config is None:
        config = get_bedrock_config()
    available = set(config.get("available_models", []))
    aliases = set(config.get("custom_aliases", {}))
    valid, invalid = [], []
    for m in models:
        base = m[len("bedrock/") :] if m.startswith("bedrock/") else m
        if base in available or base in aliases or base in BEDROCK_MODEL_MAP.values():
            valid.append(m)
        else:
            invalid.append(m)
    return valid, invalid


# ---------------------------------------------------------------------------
# Profiles (reference: providers.py:188-244; path compat SURVEY §2.5)
# ---------------------------------------------------------------------------

PROFILES_DIR = Path.home() / ".config" / "adversarial-spec" / "profiles"

_PROFILE_KEYS = ("models", "doc_type", "focus", "persona", "context", "preserve_intent")


def save_profile(name: str, settings: dict[str, Any]) -> Path:
    PROFILES_DIR.mkdir(parents=True, exist_ok=True)
    path = (PROFILES_DIR / f"{name}.json").resolve()
    if not path.is_relative_to(PROFILES_DIR.resolve()):
        raise ValueError(f"Invalid profile name: {name}")
    data = {k: settings.get(k) for k in _PROFILE_KEYS}
    path.write_text(json.dumps(data, indent=2))
    return path


def load_profile(name: str) -> Optional[dict[str, Any]]:
    path = (PROFILES_DIR / f"{name}.json").resolve()
    try:
        if not path.is_relative_to(PROFILES_DIR.resolve()):
            return None
    except ValueError:
        return None
    if not path.exists():
        return None
    try:
        return json.loads(path.read_text())
    except (json.JSONDecodeError, OSError):
        return None


def list_profiles() -> list[str]:
    if not PROFILES_DIR.exists():
        return []
    return sorted(p.stem for p in PROFILES_DIR.glob("*.json"))


# ---------------------------------------------------------------------------
# Provider discovery / default model / credential validation
# ---------------------------------------------------------------------------

_PROVIDERS: list[tuple[str, Optional[str], str]] = [
    ("OpenAI", "OPENAI_API_KEY", "gpt-4o"),
    ("Anthropic", "ANTHROPIC_API_KEY", "claude-sonnet-4-20250514"),
    ("Google", "GEMINI_API_KEY", "gemini/gemini-2.0-flash"),
    ("xAI", "XAI_API_KEY", "xai/grok-3"),
    ("Mistral", "MISTRAL_API_KEY", "mistral/mistral-large"),
    ("Groq", "GROQ_API_KEY", "groq/llama-3.3-70b-versatile"),
    ("OpenRouter", "OPENROUTER_API_KEY", "openrouter/openai/gpt-4o"),
    ("Deepseek", "DEEPSEEK_API_KEY", "deepseek/deepseek-chat"),
    ("Zhipu", "ZHIPUAI_API_KEY", "zhipu/glm-4"),
]


def local_engine_available() -> bool:
    """True when the on-node MI355X engine can run (a GPU is visible, or the
    stub arch is explicitly requested via env for CPU plumbing tests)."""
    if os.environ.get("ADVSPEC_FORCE_LOCAL"):
        return True
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


def get_available_providers() -> list[tuple[str, Optional[str], str]]:
    """(provider, env_var, default_model) for each configured provider.

    The MI355X engine is listed first when a GPU is present: on-node
    inference is this framework's primary backend.
    """
    available: list[tuple[str, Optional[str], str]] = []
    if local_engine_available():
        available.append(("MI355X local", None, "local/llama-3-8b"))
    for name, key, model in _PROVIDERS:
        if os.environ.get(key):
            available.append((name, key, model))
    if CODEX_AVAILABLE:
        available.append(("Codex CLI", None, "codex/gpt-5.2-codex"))
    if GEMINI_CLI_AVAILABLE:
        available.append(("Gemini CLI", None, "gemini-cli/gemini-3-pro-preview"))
    return available


def get_default_model() -> Optional[str]:
    """Pick a default model: local engine, then bedrock, then API keys."""
    if local_engine_available():
        cfg = get_local_config()
        models = cfg.get("available_models")
        if models:
            return f"local/{models[0]}"
        return "local/llama-3-8b"
    bedrock = get_bedrock_config()
    if bedrock.get("enabled"):
        models = bedrock.get("available_models", [])
        if models:
            return models[0]
    avail = get_available_providers()
    if avail:
        return avail[0][2]
    return None


_PROVIDER_PREFIX_MAP: dict[str, Optional[str]] = {
    "local/": None,  # on-node engine, no API key
    "gpt-": "OPENAI_API_KEY",
    "o1": "OPENAI_API_KEY",
    "claude-": "ANTHROPIC_API_KEY",
    "gemini/": "GEMINI_API_KEY",
    "xai/": "XAI_API_KEY",
    "mistral/": "MISTRAL_API_KEY",
    "groq/": "GROQ_API_KEY",
    "deepseek/": "DEEPSEEK_API_KEY",
    "zhipu/": "ZHIPUAI_API_KEY",
    "codex/": None,
    "gemini-cli/": None,
}


def validate_model_credentials(models: list[str]) -> tuple[list[str], list[str]]:
    """(valid, invalid) by credential/availability pre-flight.

    local/ models validate against the local registry + engine availability;
    CLI models against binary presence; API models against env keys; bedrock
    mode against its allow-list (reference: providers.py:418-486).
    """
    bedrock = get_bedrock_config()
    if bedrock.get("enabled"):
        return validate_bedrock_models(models, bedrock)

    valid, invalid = [], []
    for model in models:
        if model.startswith("local/"):
            try:
                resolve_local_model(model[len("local/") :])
                ok = local_engine_available()
            except ValueError:
                ok = False
            (valid if ok else invalid).append(model)
            continue
        if model.startswith("codex/"):
            (valid if CODEX_AVAILABLE else invalid).append(model)
            continue
        if model.startswith("gemini-cli/"):
            (valid if GEMINI_CLI_AVAILABLE else invalid).append(model)
            continue
        required = None
        for prefix, key in _PROVIDER_PREFIX_MAP.items():
            if model.startswith(prefix):
                required = key
                break
        else:
            valid.append(model)  # unknown scheme: let the backend decide
            continue
        if required is None or os.environ.get(required):
            valid.append(model)
        else:
            invalid.append(model)
    return valid, invalid


# ---------------------------------------------------------------------------
# Info listings
# ---------------------------------------------------------------------------

def list_providers() -> str:
    """Human-readable provider/model listing (reference: providers.py:247-334)."""
    lines = ["", "=== Available Providers ===", ""]
    avail = get_available_providers()
    if not avail:
        lines.append("No providers configured.")
        lines.append("")
        lines.append("Set an API key (e.g. OPENAI_API_KEY), install the codex/gemini")
        lines.append("CLI, or run on a machine with MI355X GPUs for local inference.")
    else:
        for name, key, model in avail:
            via = key if key else "no API key needed"
            lines.append(f"  {name:<14} default: {model:<34} ({via})")
    lines.append("")
    lines.append("=== Local (MI355X) model registry ===")
    cfg = get_local_config()
    names = sorted(set(LOCAL_MODEL_MAP) | set(cfg.get("custom_aliases", {})))
    for n in names:
        try:
            spec = resolve_local_model(n, cfg)
            w = spec.get("weights") or "random-init"
            lines.append(f"  local/{n:<14} arch={spec['arch']:<12} weights={w}")
        except ValueError:
            continue
    bedrock = get_bedrock_config()
    lines.append("")
    if bedrock.get("enabled"):
        lines.append(f"=== Bedrock (enabled, region {bedrock.get('region', '?')}) ===")
        for m in bedrock.get("available_models", []):
            lines.append(f"  {m}")
    else:
        lines.append("Bedrock: disabled (enable with: debate.py bedrock enable --region <r>)")
    lines.append("")
    lines.append("=== Focus areas ===")
    lines.append("  " + ", ".join(FOCUS_AREAS))
    lines.append("=== Personas ===")
    lines.append("  " + ", ".join(PERSONAS))
    return "\n".join(lines)


# ---------------------------------------------------------------------------
# bedrock / local subcommands (reference: providers.py:489-656)
# ---------------------------------------------------------------------------

def handle_bedrock_command(subcommand: Optional[str], arg: Optional[str],
                           extra: Optional[str], region: Optional[str]) -> int:
    """`debate.py bedrock {status,enable,disable,add-model,remove-model,alias,list-models}`.

    Returns the process exit code (0 ok, 1 usage/processing error).
    """
    config = load_global_config()
    bedrock = config.setdefault(
        "bedrock",
        {"enabled": False, "region": None, "available_models": [], "custom_aliases": {}},
    )

    if subcommand in (None, "status"):
        print("Bedrock mode:", "enabled" if bedrock.get("enabled") else "disabled")
        print("Region:", bedrock.get("region") or "(not set)")
        print("Available models:", ", ".join(bedrock.get("available_models", [])) or "(none)")
        aliases = bedrock.get("custom_aliases", {})
        if aliases:
            print("Custom aliases:")
            for k, v in aliases.items():
                print(f"  {k} -> {v}")
        return 0
    if subcommand == "enable":
        if region:
            bedrock["region"] = region
        if not bedrock.get("region"):
            print("Error: --region is required to enable Bedrock", file=sys.stderr)
            return 1
        bedrock["enabled"] = True
        save_global_config(config)
        print(f"Bedrock mode enabled (region {bedrock['region']})")
        return 0
    if subcommand == "disable":
        bedrock["enabled"] = False
        save_global_config(config)
        print("Bedrock mode disabled")
        return 0
    if subcommand == "add-model":
        if not arg:
            print("Error: model name required", file=sys.stderr)
            return 1
        if arg not in bedrock["available_models"]:
            bedrock["available_models"].append(arg)
        save_global_config(config)
        print(f"Added Bedrock model: {arg} -> {resolve_bedrock_model(arg, bedrock)}")
        return 0
    if subcommand == "remove-model":
        if not arg:
            print("Error: model name required", file=sys.stderr)
            return 1
        if arg in bedrock["available_models"]:
            bedrock["available_models"].remove(arg)
            save_global_config(config)
            print(f"Removed Bedrock model: {arg}")
        else:
            print(f"Model not in list: {arg}", file=sys.stderr)
            return 1
        return 0
    if subcommand == "alias":
        if not arg or not extra:
            print("Error: usage: bedrock alias <name> <bedrock-model-id>", file=sys.stderr)
            return 1
        bedrock.setdefault("custom_aliases", {})[arg] = extra
        save_global_config(config)
        print(f"Alias added: {arg} -> {extra}")
        return 0
    if subcommand == "list-models":
        print("Friendly name map:")
        for k, v in BEDROCK_MODEL_MAP.items():
            print(f"  {k:<22} {v}")
        for k, v in bedrock.get("custom_aliases", {}).items():
            print(f"  {k:<22} {v}  (alias)")
        return 0
    print(f"Unknown bedrock subcommand: {subcommand}", file=sys.stderr)
    return 1


def handle_local_command(subcommand: Optional[str], arg: Optional[str],
                         extra: Optional[str], weights: Optional[str],
                         gpu: Optional[int],
                         dtype: Optional[str] = None) -> int:
    """`debate.py local {status,add-model,remove-model,alias,list-models}` —
    manages the MI355X model registry section of the global config."""
    config = load_global_config()
    local = config.setdefault(
        "local", {"enabled": True, "available_models": [], "custom_aliases": {}}
    )

    if subcommand in (None, "status"):
        print("Local MI355X engine:", "available" if local_engine_available() else "no GPU visible")
        print("Registered models:", ", ".join(local.get("available_models", [])) or "(none)")
        aliases = local.get("custom_aliases", {})
        if aliases:
            print("Custom aliases:")
            for k, v in aliases.items():
                print(f"  {k} -> {json.dumps(v)}")
        return 0
    if subcommand == "add-model":
        if not arg:
            print("Error: model name required", file=sys.stderr)
            return 1
        try:
            resolve_local_model(arg, local)
        except ValueError as e:
            print(f"Error: {e}", file=sys.stderr)
            return 1
        if arg not in local["available_models"]:
            local["available_models"].append(arg)
        save_global_config(config)
        print(f"Added local model: {arg}")
        return 0
    if subcommand == "remove-model":
        if not arg:
            print("Error: model name required", file=sys.stderr)
            return 1
        if arg in local["available_models"]:
            local["available_models"].remove(arg)
            save_global_config(config)
            print(f"Removed local model: {arg}")
            return 0
        print(f"Model not in list: {arg}", file=sys.stderr)
        return 1
    if subcommand == "alias":
        if not arg or not extra:
            print("Error: usage: local alias <name> <arch> [--weights PATH] [--gpu N]",
                  file=sys.stderr)
            return 1
        if extra not in LOCAL_MODEL_MAP and extra not in LOCAL_MODEL_MAP.values():
            print(f"Error: unknown arch '{extra}'", file=sys.stderr)
            return 1
        spec: dict[str, Any] = {"arch": LOCAL_MODEL_MAP.get(extra, extra)}
        if weights:
            spec["weights"] = weights
        if gpu is not None:
            spec["gpu"] = gpu
        if dtype:
            spec["dtype"] = dtype
        local.setdefault("custom_aliases", {})[arg] = spec
        save_global_config(config)
        print(f"Local alias added: {arg} -> {json.dumps(spec)}")
        return 0
    if subcommand == "list-models":
        for k, v in LOCAL_MODEL_MAP.items():
            print(f"  {k:<16} arch={v}")
        for k, v in local.get("custom_aliases", {}).items():
            print(f"  {k:<16} {json.dumps(v)}  (alias)")
        return 0
    print(f"Unknown local subcommand: {subcommand}", file=sys.stderr)
    return 1
