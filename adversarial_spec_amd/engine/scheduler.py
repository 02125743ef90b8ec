"""Round scheduler: prompt assembly, opponent fan-out, retries, consensus.

The reference's `call_models_parallel` is a ThreadPoolExecutor over HTTPS
calls (models.py:681-722). Here the same API fans out over opponent
backends; `local/` opponents are pinned to GPUs round-robin, so an N-GPU
node runs N opponents genuinely in parallel (the "opponent parallelism"
axis). Results always come back in MODEL-LIST ORDER —
deterministic, fixing the reference's arrival-order nondeterminism in
"latest spec wins" (reference: debate.py:859-863).

The distributed (one-process-per-GPU, RCCL all-gather) round path lives in
parallel/consensus.py and is used by bench.py / torchrun launches; this
module is the single-invocation CLI path.
"""

from __future__ import annotations

import sys
import time
from concurrent.futures import ThreadPoolExecutor
from typing import Optional

from ..prompts import (
    FOCUS_AREAS,
    PRESERVE_INTENT_PROMPT,
    PRESS_PROMPT_TEMPLATE,
    REVIEW_PROMPT_TEMPLATE,
    get_doc_type_name,
    get_system_prompt,
)
from ..protocol import (
    MAX_RETRIES,
    RETRY_BASE_DELAY,
    ModelResponse,
    cost_tracker,
    detect_agreement,
    extract_spec,
)
from ..providers import DEFAULT_CODEX_REASONING
from .backend import LocalBackend, get_backend


def load_context_files(context_paths: list[str]) -> str:
    """Format --context files into a prompt section (reference: models.py:130-146)."""
    if not context_paths:
        return ""
    sections = []
    for path in context_paths:
        try:
            from pathlib import Path

            content = Path(path).read_text()
            sections.append(f"### Context: {path}\n```\n{content}\n```")
        except Exception as e:
            sections.append(f"### Context: {path}\n[Error loading file: {e}]")
    return (
        "## Additional Context\nThe following documents are provided as context:\n\n"
        + "\n\n".join(sections)
    )


def build_user_message(
    spec: str,
    round_num: int,
    doc_type: str,
    press: bool = False,
    focus: Optional[str] = None,
    context: Optional[str] = None,
    preserve_intent: bool = False,
) -> str:
    """Assemble the per-round user prompt (reference: models.py:482-503)."""
    doc_type_name = get_doc_type_name(doc_type)
    focus_section = ""
    if focus and focus.lower() in FOCUS_AREAS:
        focus_section = FOCUS_AREAS[focus.lower()]
    elif focus:
        focus_section = (
            f"**CRITICAL FOCUS: {focus.upper()}**\n"
            f"Prioritize analysis of {focus} concerns above all else."
        )
    if preserve_intent:
        focus_section = PRESERVE_INTENT_PROMPT + "\n\n" + focus_section
    template = PRESS_PROMPT_TEMPLATE if press else REVIEW_PROMPT_TEMPLATE
    return template.format(
        round=round_num,
        doc_type_name=doc_type_name,
        spec=spec,
        focus_section=focus_section,
        context_section=context or "",
    )


def call_single_model(
    model: str,
    spec: str,
    round_num: int,
    doc_type: str,
    press: bool = False,
    focus: Optional[str] = None,
    persona: Optional[str] = None,
    context: Optional[str] = None,
    preserve_intent: bool = False,
    codex_reasoning: str = DEFAULT_CODEX_REASONING,
    codex_search: bool = False,
    timeout: int = 600,
    bedrock_mode: bool = False,
    bedrock_region: Optional[str] = None,
    device: Optional[str] = None,
    max_tokens: int = 8000,
    temperature: float = 0.7,
) -> ModelResponse:
    """One opponent's critique with retry/backoff and protocol parsing.

    Retry policy: MAX_RETRIES attempts, exponential backoff 1s/2s/4s
    (reference: models.py:46-47, 611-674). A GPU fault (HIP error / OOM)
    retries the same way a remote 5xx would; final failure returns
    ModelResponse(error=...) so the round continues with survivors.
    """
    system_prompt = get_system_prompt(doc_type, persona)
    user_message = build_user_message(
        spec, round_num, doc_type, press, focus, context, preserve_intent
    )
    backend = get_backend(
        model,
        device=device,
        codex_reasoning=codex_reasoning,
        codex_search=codex_search,
        bedrock_mode=bedrock_mode,
        bedrock_region=bedrock_region,
    )
    # o-series models reject custom temperature; litellm backend also guards
    # itself, but sampling knobs are decided here for local opponents too.
    temp = temperature

    last_error: Optional[Exception] = None
    for attempt in range(MAX_RETRIES):
        try:
            content, in_tok, out_tok = backend.generate(
                system_prompt,
                user_message,
                max_tokens=max_tokens,
                temperature=temp,
                timeout=timeout,
            )
            agreed = detect_agreement(content)
            spec_out = extract_spec(content)
            if not agreed and not spec_out:
                print(
                    f"Warning: {model} provided critique but no [SPEC] tags found. "
                    "Response may be malformed.",
                    file=sys.stderr,
                )
            cost = cost_tracker.add(model, in_tok, out_tok)
            timings = getattr(backend, "last_timings", {})
            return ModelResponse(
                model=model,
                response=content,
                agreed=agreed,
                spec=spec_out,
                input_tokens=in_tok,
                output_tokens=out_tok,
                cost=cost,
                timings=dict(timings) if timings else {},
            )
        except Exception as e:
            last_error = e
            if attempt < MAX_RETRIES - 1:
                time.sleep(RETRY_BASE_DELAY * (2**attempt))

    msg = str(last_error)
    # Bedrock error translation (reference: models.py:655-661)
    if "AccessDenied" in msg:
        msg = f"Model not enabled in Bedrock console: {model}"
    elif "ValidationException" in msg and bedrock_mode:
        msg = f"Invalid Bedrock model ID: {model}"
    return ModelResponse(model=model, error=msg)


def _assign_devices(models: list[str]) -> list[Optional[str]]:
    """Pin local/ opponents to GPUs round-robin (one opponent per GPU when
    GPUs suffice; co-residency in 288 GB HBM3E otherwise)."""
    try:
        import torch

        ngpu = torch.cuda.device_count() if torch.cuda.is_available() else 0
    except Exception:
        ngpu = 0
    out: list[Optional[str]] = []
    local_idx = 0
    for m in models:
        if m.startswith("local/") and ngpu > 0:
            out.append(f"cuda:{local_idx % ngpu}")
            local_idx += 1
        else:
            out.append(None)
    return out


def call_models_parallel(
    models: list[str],
    spec: str,
    round_num: int,
    doc_type: str,
    press: bool = False,
    focus: Optional[str] = None,
    persona: Optional[str] = None,
    context: Optional[str] = None,
    preserve_intent: bool = False,
    codex_reasoning: str = DEFAULT_CODEX_REASONING,
    codex_search: bool = False,
    timeout: int = 600,
    bedrock_mode: bool = False,
    bedrock_region: Optional[str] = None,
) -> list[ModelResponse]:
    """Fan the identical prompt out to every opponent; gather all results.

    Returns results in model-list order (deterministic consensus and
    "latest spec" selection downstream).
    """
    devices = _assign_devices(models)
    with ThreadPoolExecutor(max_workers=max(1, len(models))) as pool:
        futures = [
            pool.submit(
                call_single_model,
                m,
                spec,
                round_num,
                doc_type,
                press,
                focus,
                persona,
                context,
                preserve_intent,
                codex_reasoning,
                codex_search,
                timeout,
                bedrock_mode,
                bedrock_region,
                devices[i],
            )
            for i, m in enumerate(models)
        ]
        return [f.result() for f in futures]
