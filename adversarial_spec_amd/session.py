"""Session state and per-round checkpoints (reference: session.py).

Two host-side persistence mechanisms, path-compatible with
the reference:
  1. session JSON at ~/.config/adversarial-spec/sessions/<id>.json
  2. per-round spec checkpoints at ./.adversarial-spec-checkpoints/
     [<session>-]round-<n>.md

GPU state (weights, KV caches) is deliberately NOT checkpointed: a debate
round is re-playable from the spec text, and on MI355X re-prefill of even a
32k spec is cheaper than serializing hundreds of GB of HBM3E-resident state.
"""

from __future__ import annotations

import json
from dataclasses import asdict, dataclass, field
from datetime import datetime
from pathlib import Path
from typing import Any, Optional

SESSIONS_DIR = Path.home() / ".config" / "adversarial-spec" / "sessions"
CHECKPOINTS_DIR = Path(".adversarial-spec-checkpoints")


@dataclass
class SessionState:
    """Debate session persisted between CLI invocations (reference: session.py:16-71)."""

    session_id: str
    spec: str
    round: int = 1
    doc_type: str = "tech"
    models: list[str] = field(default_factory=list)
    focus: Optional[str] = None
    persona: Optional[str] = None
    preserve_intent: bool = False
    created_at: str = ""
    updated_at: str = ""
    history: list[dict[str, Any]] = field(default_factory=list)

    def _path(self) -> Path:
        path = (SESSIONS_DIR / f"{self.session_id}.json").resolve()
        if not path.is_relative_to(SESSIONS_DIR.resolve()):
            raise ValueError(f"Invalid session ID: {self.session_id}")
        return path

    def save(self) -> None:
        SESSIONS_DIR.mkdir(parents=True, exist_ok=True)
        self.updated_at = datetime.now().isoformat()
        self._path().write_text(json.dumps(asdict(self), indent=2))

    @classmethod
    def load(cls, session_id: str) -> "SessionState":
        path = (SESSIONS_DIR / f"{session_id}.json").resolve()
        if not path.is_relative_to(SESSIONS_DIR.resolve()):
            raise FileNotFoundError(f"Invalid session ID: {session_id}")
        if not path.exists():
            raise FileNotFoundError(f"Session not found: {session_id}")
        data = json.loads(path.read_text())
        known = {f for f in cls.__dataclass_fields__}  # tolerate newer fields
        return cls(**{k: v for k, v in data.items() if k in known})

    @classmethod
    def list_sessions(cls) -> list[dict[str, Any]]:
        """Summaries of saved sessions, newest first (reference: session.py:52-71)."""
        if not SESSIONS_DIR.exists():
            return []
        out = []
        for p in SESSIONS_DIR.glob("*.json"):
            try:
                d = json.loads(p.read_text())
                out.append(
                    {
                        "session_id": d.get("session_id", p.stem),
                        "round": d.get("round", 1),
                        "doc_type": d.get("doc_type", "tech"),
                        "models": d.get("models", []),
                        "updated_at": d.get("updated_at", ""),
                        "rounds_completed": len(d.get("history", [])),
                    }
                )
            except (json.JSONDecodeError, OSError):
                continue
        out.sort(key=lambda s: s.get("updated_at", ""), reverse=True)
        return out


def save_checkpoint(spec: str, round_num: int, session_id: Optional[str] = None) -> Path:
    """Write the round's spec to ./.adversarial-spec-checkpoints/ for manual
    rollback (reference: session.py:74-82)."""
    CHECKPOINTS_DIR.mkdir(parents=True, exist_ok=True)
    name = f"{session_id}-round-{round_num}.md" if session_id else f"round-{round_num}.md"
    path = (CHECKPOINTS_DIR / name).resolve()
    if not path.is_relative_to(CHECKPOINTS_DIR.resolve()):
        raise ValueError(f"Invalid session ID: {session_id}")
    path.write_text(spec)
    return path
