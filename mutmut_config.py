"""Mutation-testing hook (mutmut): skip mutants that no test should kill.

The reference treats mutation testing as first-class; this
hook mirrors its policy: don't mutate logging strings, static config data
tables, or docstrings — those mutants are noise, not missed coverage.
"""

from __future__ import annotations

_SKIP_SUBSTRINGS = (
    "print(",            # progress/log lines
    "file=sys.stderr",
    "description=",      # argparse help surface
    "help=",
)

_SKIP_FILES = (
    "prompts.py",        # prompt prose: mutants are wording changes
    "config.py",         # static architecture tables
)


def pre_mutation(context) -> None:
    line = (context.current_source_line or "").strip()
    if any(s in line for s in _SKIP_SUBSTRINGS):
        context.skip = True
        return
    filename = getattr(context, "filename", "") or ""
    if any(filename.endswith(f) for f in _SKIP_FILES):
        context.skip = True
