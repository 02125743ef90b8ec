"""Wire-format protocol: agreement markers, spec/task extraction, cost tracking.

Preserves the reference's compat surface exactly:
  - [AGREE] substring anywhere in a response marks agreement
    (reference: models.py:149-151)
  - spec text is the region between the first [SPEC] and the first [/SPEC],
    stripped; None when either tag is absent (reference: models.py:154-160)
  - [TASK]...[/TASK] blocks with title/type/priority/description/
    acceptance_criteria fields, multi-line continuation, `- ` list items,
    titleless tasks dropped (reference: models.py:163-247)
  - unified diff with fromfile="previous"/tofile="current"
    (reference: models.py:263-271)

The CostTracker is thread- and stream-safe by construction (a lock guards
every mutation), fixing the reference's unlocked `+=` from worker threads
(reference: models.py:127).
"""

from __future__ import annotations

import difflib
import threading
from dataclasses import dataclass, field
from typing import Optional

AGREE_MARKER = "[AGREE]"
SPEC_OPEN = "[SPEC]"
SPEC_CLOSE = "[/SPEC]"
TASK_OPEN = "[TASK]"
TASK_CLOSE = "[/TASK]"

# Retry policy shared by every backend (reference: models.py:46-47).
MAX_RETRIES = 3
RETRY_BASE_DELAY = 1.0  # seconds; attempt k sleeps RETRY_BASE_DELAY * 2**k


@dataclass
class ModelResponse:
    """Result of one opponent's critique (reference: models.py:67-78)."""

    model: str
    response: str = ""
    agreed: bool = False
    spec: Optional[str] = None
    error: Optional[str] = None
    input_tokens: int = 0
    output_tokens: int = 0
    cost: float = 0.0
    # MI355X additions (not in the reference's JSON schema unless requested):
    # per-phase wall-clock for observability.
    timings: dict = field(default_factory=dict)


def detect_agreement(response: str) -> bool:
    """True when the response contains the literal agreement marker."""
    return AGREE_MARKER in response


def extract_spec(response: str) -> Optional[str]:
    """Return text between the first [SPEC] and first [/SPEC], stripped."""
    if SPEC_OPEN not in response or SPEC_CLOSE not in response:
        return None
    start = response.find(SPEC_OPEN) + len(SPEC_OPEN)
    end = response.find(SPEC_CLOSE)
    return response[start:end].strip()


_TASK_FIELDS = ("title", "type", "priority", "description", "acceptance_criteria")


def extract_tasks(response: str) -> list[dict]:
    """Parse [TASK] blocks into dicts (line-oriented grammar, see module doc).

    Grammar (reference contract, prompts.py:270-279 / models.py:163-247):
    each field starts a `key:` line; later non-field lines continue the
    current field; under acceptance_criteria only `- ` lines append items;
    a task without a title is dropped.
    """
    tasks: list[dict] = []
    for chunk in response.split(TASK_OPEN)[1:]:
        if TASK_CLOSE not in chunk:
            continue
        body = chunk.split(TASK_CLOSE)[0].strip()
        task: dict = {}
        key: Optional[str] = None
        buf: list[str] = []

        def flush() -> None:
            if key is None:
                return
            if key == "acceptance_criteria":
                task[key] = list(buf)
            else:
                task[key] = "\n".join(buf).strip() if len(buf) > 1 else (buf[0] if buf else "")

        for raw in body.split("\n"):
            line = raw.strip()
            matched = None
            for f in _TASK_FIELDS:
                if line.startswith(f + ":"):
                    matched = f
                    break
            if matched is not None:
                flush()
                key = matched
                rest = line[len(matched) + 1 :].strip()
                buf = [] if matched == "acceptance_criteria" else [rest]
            elif line.startswith("- ") and key == "acceptance_criteria":
                buf.append(line[2:])
            elif key is not None:
                buf.append(line)
        # final field: multi-line fields join ALL lines (matches reference tail
        # handling at models.py:236-241)
        if key is not None:
            if key == "acceptance_criteria":
                task[key] = list(buf)
            else:
                task[key] = "\n".join(buf).strip()

        if task.get("title"):
            tasks.append(task)
    return tasks


def get_critique_summary(response: str, max_length: int = 300) -> str:
    """Critique text before the [SPEC] tag, truncated with an ellipsis."""
    spec_start = response.find(SPEC_OPEN)
    critique = response[:spec_start].strip() if spec_start > 0 else response
    if len(critique) > max_length:
        critique = critique[:max_length] + "..."
    return critique


def generate_diff(previous: str, current: str) -> str:
    """Unified diff between two spec versions (reference: models.py:263-271)."""
    diff = difflib.unified_diff(
        previous.splitlines(keepends=True),
        current.splitlines(keepends=True),
        fromfile="previous",
        tofile="current",
    )
    return "".join(diff)


class CostTracker:
    """Token/cost accumulator with per-model breakdown.

    Same reporting surface as the reference (models.py:81-123) but
    explicitly lock-guarded: backends may report from worker threads or
    stream callbacks concurrently.
    """

    def __init__(self) -> None:
        self._lock = threading.Lock()
        self.total_cost = 0.0
        self.total_input_tokens = 0
        self.total_output_tokens = 0
        self.by_model: dict[str, dict] = {}

    def add(self, model: str, input_tokens: int, output_tokens: int) -> float:
        from .providers import get_model_cost

        rates = get_model_cost(model)
        cost = (
            input_tokens / 1_000_000 * rates["input"]
            + output_tokens / 1_000_000 * rates["output"]
        )
        with self._lock:
            self.total_cost += cost
            self.total_input_tokens += input_tokens
            self.total_output_tokens += output_tokens
            entry = self.by_model.setdefault(
                model, {"cost": 0.0, "input_tokens": 0, "output_tokens": 0}
            )
            entry["cost"] += cost
            entry["input_tokens"] += input_tokens
            entry["output_tokens"] += output_tokens
        return cost

    def reset(self) -> None:
        with self._lock:
            self.total_cost = 0.0
            self.total_input_tokens = 0
            self.total_output_tokens = 0
            self.by_model = {}

    def summary(self) -> str:
        with self._lock:
            lines = [
                "",
                "=== Cost Summary ===",
                f"Total input tokens:  {self.total_input_tokens:,}",
                f"Total output tokens: {self.total_output_tokens:,}",
                f"Total cost:          ${self.total_cost:.4f}",
            ]
            if self.by_model:
                lines.append("")
                lines.append("By model:")
                for model, e in self.by_model.items():
                    lines.append(
                        f"  {model}: ${e['cost']:.4f} "
                        f"({e['input_tokens']:,} in / {e['output_tokens']:,} out)"
                    )
        return "\n".join(lines)


# Global tracker, mirroring the reference's module-level singleton
# (models.py:127) so the CLI can report a whole-invocation total.
cost_tracker = CostTracker()
