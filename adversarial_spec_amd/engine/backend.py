"""Opponent backends: local MI355X engine, stub, litellm API, CLI subprocesses.

All backends implement one contract:

    generate(system_prompt, user_message, *, max_tokens, temperature,
             timeout, **knobs) -> (text, input_tokens, output_tokens)

The scheduler (engine/scheduler.py) picks a backend from the model-string
prefix (SURVEY.md §2.5 routing): `local/` is the on-node engine (this
framework's L1), `stub/` a deterministic fake (BASELINE config 1 and the
test suite's mock backend, mirroring the reference's patched `completion`),
`codex/` and `gemini-cli/` subprocess CLIs (reference: models.py:274-454),
everything else litellm over HTTPS (reference: models.py:614-628).
"""

from __future__ import annotations

import json
import os
import subprocess
import threading
from typing import Optional

from ..protocol import AGREE_MARKER, SPEC_CLOSE, SPEC_OPEN
from ..providers import (
    CODEX_AVAILABLE,
    DEFAULT_CODEX_REASONING,
    GEMINI_CLI_AVAILABLE,
    resolve_local_model,
)


def is_o_series_model(model: str) -> bool:
    """o-series reasoning models reject custom temperature
    (reference: models.py:50-64): `o1`, `*/o1*`, `*-o1*`, case-insensitive."""
    m = model.lower()
    base = m.split("/")[-1]
    return base.startswith("o1") or "-o1" in base


class StubBackend:
    """Deterministic canned opponent (no GPU, no network).

    Model-string dialects under the `stub/` prefix:
      stub/agree     -> always [AGREE] + [SPEC]
      stub/critique  -> always critiques with a revised [SPEC]
      stub/error     -> always raises (fault-isolation paths)
      stub/flaky     -> fails twice per process, then succeeds (retry paths)
      stub/<other>   -> critiques on round 1, agrees from round 2
    """

    _flaky_counts: dict[str, int] = {}
    _flaky_lock = threading.Lock()

    def __init__(self, model: str) -> None:
        self.model = model
        self.mode = model.split("/", 1)[1] if "/" in model else "auto"

    @staticmethod
    def _round_from_message(user_message: str) -> int:
        # REVIEW/PRESS templates open with "This is round {n} of ..."
        head = user_message[:64]
        for tokenised in head.replace("\n", " ").split(" "):
            if tokenised.isdigit():
                return int(tokenised)
        return 1

    def generate(self, system_prompt: str, user_message: str, *, max_tokens: int,
                 temperature: float, timeout: float, **_) -> tuple[str, int, int]:
        if self.mode == "error":
            raise RuntimeError("stub backend: simulated failure")
        if self.mode == "flaky":
            with self._flaky_lock:
                n = self._flaky_counts.get(self.model, 0)
                self._flaky_counts[self.model] = n + 1
            if n < 2:
                raise RuntimeError(f"stub backend: simulated transient failure {n + 1}")
        rnd = self._round_from_message(user_message)
        agree = self.mode == "agree" or (self.mode not in ("critique",) and rnd >= 2)
        revised = (
            "# Revised Specification (stub)\n\n"
            f"Round {rnd} revision produced by {self.model}.\n"
            "## Error Handling Strategy\nAdded explicit failure modes.\n"
        )
        if agree:
            text = f"{AGREE_MARKER}\n{SPEC_OPEN}\n{revised}{SPEC_CLOSE}\n"
        else:
            text = (
                f"The document lacks an error-handling section (round {rnd}).\n"
                "1. No failure modes are enumerated.\n"
                "2. Performance targets carry no numbers.\n\n"
                f"{SPEC_OPEN}\n{revised}{SPEC_CLOSE}\n"
            )
        in_tok = (len(system_prompt) + len(user_message)) // 4
        return text, in_tok, len(text) // 4

    @classmethod
    def reset_flaky(cls) -> None:
        with cls._flaky_lock:
            cls._flaky_counts.clear()


class LocalBackend:
    """On-node MI355X inference for a `local/<name>` model."""

    def __init__(self, model: str, device: Optional[str] = None) -> None:
        self.model = model
        name = model.split("/", 1)[1]
        self.spec = resolve_local_model(name)
        self.device = device
        self.last_timings: dict[str, float] = {}

    def generate(self, system_prompt: str, user_message: str, *, max_tokens: int,
                 temperature: float, timeout: float, top_p: float = 1.0, **_):
        from .local import get_engine

        engine = get_engine(self.spec, device=self.device)
        text, in_tok, out_tok, timings = engine.generate(
            system_prompt, user_message, max_tokens=max_tokens,
            temperature=temperature, timeout=timeout, top_p=top_p,
        )
        self.last_timings = timings
        return text, in_tok, out_tok


class LiteLLMBackend:
    """Remote API opponent via litellm (optional dependency).

    Kept for full parity with the reference's provider surface; in an
    air-gapped MI355X deployment this backend simply reports its absence
    and the round continues with local opponents (per-opponent fault
    isolation).
    """

    def __init__(self, model: str, bedrock_mode: bool = False,
                 bedrock_region: Optional[str] = None) -> None:
        self.model = model
        self.bedrock_mode = bedrock_mode
        self.bedrock_region = bedrock_region

    def generate(self, system_prompt: str, user_message: str, *, max_tokens: int,
                 temperature: float, timeout: float, **_):
        try:
            from litellm import completion  # type: ignore
        except ImportError as e:
            raise RuntimeError(
                "litellm is not installed; remote API opponents are unavailable "
                "on this node (local/ opponents run on-GPU without it)"
            ) from e
        actual = self.model
        if self.bedrock_mode:
            if self.bedrock_region:
                os.environ["AWS_REGION"] = self.bedrock_region
            if not actual.startswith("bedrock/"):
                actual = f"bedrock/{actual}"
        kwargs = {
            "model": actual,
            "messages": [
                {"role": "system", "content": system_prompt},
                {"role": "user", "content": user_message},
            ],
            "max_tokens": max_tokens,
            "timeout": timeout,
        }
        if not is_o_series_model(self.model):
            kwargs["temperature"] = temperature
        resp = completion(**kwargs)
        content = resp.choices[0].message.content or ""
        usage = getattr(resp, "usage", None)
        in_tok = getattr(usage, "prompt_tokens", 0) if usage else 0
        out_tok = getattr(usage, "completion_tokens", 0) if usage else 0
        return content, in_tok, out_tok


def _combined_prompt(system_prompt: str, user_message: str) -> str:
    return f"SYSTEM INSTRUCTIONS:\n{system_prompt}\n\nUSER REQUEST:\n{user_message}"


class CodexCLIBackend:
    """`codex exec --json` subprocess opponent (reference: models.py:274-370)."""

    def __init__(self, model: str, reasoning: str = DEFAULT_CODEX_REASONING,
                 search: bool = False) -> None:
        self.model = model
        self.reasoning = reasoning
        self.search = search

    def generate(self, system_prompt: str, user_message: str, *, max_tokens: int,
                 temperature: float, timeout: float, **_):
        if not CODEX_AVAILABLE:
            raise RuntimeError("Codex CLI not found in PATH")
        actual = self.model.split("/", 1)[1] if "/" in self.model else self.model
        cmd = [
            "codex", "exec", "--json", "--full-auto", "--skip-git-repo-check",
            "--model", actual, "-c", f'model_reasoning_effort="{self.reasoning}"',
        ]
        if self.search:
            cmd.append("--search")
        cmd.append(_combined_prompt(system_prompt, user_message))
        try:
            result = subprocess.run(cmd, capture_output=True, text=True, timeout=timeout)
        except subprocess.TimeoutExpired as e:
            raise RuntimeError(f"Codex CLI timed out after {timeout}s") from e
        except FileNotFoundError as e:
            raise RuntimeError("Codex CLI not found in PATH") from e
        if result.returncode != 0:
            msg = result.stderr.strip() or f"exit code {result.returncode}"
            raise RuntimeError(f"Codex CLI failed: {msg}")
        text, in_tok, out_tok = "", 0, 0
        for line in result.stdout.strip().split("\n"):
            line = line.strip()
            if not line:
                continue
            try:
                event = json.loads(line)
            except json.JSONDecodeError:
                continue
            if event.get("type") == "item.completed":
                item = event.get("item", {})
                if item.get("type") == "agent_message":
                    text = item.get("text", "")
            elif event.get("type") == "turn.completed":
                usage = event.get("usage", {})
                in_tok = usage.get("input_tokens", 0)
                out_tok = usage.get("output_tokens", 0)
        if not text:
            raise RuntimeError("No agent message found in Codex output")
        return text, in_tok, out_tok


class GeminiCLIBackend:
    """`gemini -m <model> -y` subprocess opponent (reference: models.py:373-454)."""

    _NOISE_PREFIXES = ("Loaded cached", "Server ", "Loading extension")

    def __init__(self, model: str) -> None:
        self.model = model

    def generate(self, system_prompt: str, user_message: str, *, max_tokens: int,
                 temperature: float, timeout: float, **_):
        if not GEMINI_CLI_AVAILABLE:
            raise RuntimeError("Gemini CLI not found in PATH")
        actual = self.model.split("/", 1)[1] if "/" in self.model else self.model
        prompt = _combined_prompt(system_prompt, user_message)
        try:
            result = subprocess.run(
                ["gemini", "-m", actual, "-y"], input=prompt,
                capture_output=True, text=True, timeout=timeout,
            )
        except subprocess.TimeoutExpired as e:
            raise RuntimeError(f"Gemini CLI timed out after {timeout}s") from e
        except FileNotFoundError as e:
            raise RuntimeError("Gemini CLI not found in PATH") from e
        if result.returncode != 0:
            msg = result.stderr.strip() or f"exit code {result.returncode}"
            raise RuntimeError(f"Gemini CLI failed: {msg}")
        lines = [
            ln for ln in result.stdout.strip().split("\n")
            if not any(ln.startswith(p) for p in self._NOISE_PREFIXES)
        ]
        text = "\n".join(lines).strip()
        if not text:
            raise RuntimeError("No response from Gemini CLI")
        # CLI reports no usage; estimate 4 chars/token (reference: models.py:444-447)
        return text, len(prompt) // 4, len(text) // 4


def get_backend(model: str, *, device: Optional[str] = None,
                codex_reasoning: str = DEFAULT_CODEX_REASONING,
                codex_search: bool = False, bedrock_mode: bool = False,
                bedrock_region: Optional[str] = None):
    """Route a model string to its backend (prefix rules)."""
    if os.environ.get("ADVSPEC_BACKEND") == "stub":
        return StubBackend(model)
    if model.startswith("stub/") or model == "stub":
        return StubBackend(model)
    if model.startswith("local/"):
        return LocalBackend(model, device=device)
    if model.startswith("codex/"):
        return CodexCLIBackend(model, reasoning=codex_reasoning, search=codex_search)
    if model.startswith("gemini-cli/"):
        return GeminiCLIBackend(model)
    return LiteLLMBackend(model, bedrock_mode=bedrock_mode, bedrock_region=bedrock_region)
