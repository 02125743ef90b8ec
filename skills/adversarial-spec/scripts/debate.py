#!/usr/bin/env python3
"""Skill-facing shim: `python3 scripts/debate.py ...` == the package CLI.

The /adversarial-spec skill invokes this path via the Bash tool
(reference layout: skills/adversarial-spec/scripts/debate.py).
"""

import sys
from pathlib import Path

_REPO_ROOT = Path(__file__).resolve().parents[3]
if str(_REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(_REPO_ROOT))

from adversarial_spec_amd.cli.debate import main  # noqa: E402

if __name__ == "__main__":
    sys.exit(main())
