"""`debate.py` — CLI entry for the adversarial spec debate.

Preserves the reference CLI surface (reference: debate.py:397-432):
actions {critique, providers, send-final, diff, export-tasks, focus-areas,
personas, profiles, save-profile, sessions, bedrock} plus the MI355X
registry action `local`; same flags, stdin=spec, stdout=text-or-JSON,
exit codes 0 (success) / 1 (processing error) / 2 (missing key/config).

JSON output schema is byte-compatible with the reference
(reference: debate.py:909-941).
"""

from __future__ import annotations

import argparse
import json
import sys
from typing import Any, Optional

from .. import protocol, providers, telegram
from ..engine.scheduler import call_models_parallel, load_context_files
from ..prompts import (
    EXPORT_TASKS_PROMPT,
    FOCUS_AREAS,
    PERSONAS,
    get_doc_type_name,
)
from ..protocol import ModelResponse, cost_tracker, extract_tasks, generate_diff
from ..providers import DEFAULT_CODEX_REASONING
from ..session import SessionState, save_checkpoint

EXIT_OK = 0
EXIT_ERROR = 1
EXIT_CONFIG = 2

ACTIONS = [
    "critique",
    "providers",
    "send-final",
    "diff",
    "export-tasks",
    "focus-areas",
    "personas",
    "profiles",
    "save-profile",
    "sessions",
    "bedrock",
    "local",
    "serve",
]


def create_parser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser(
        prog="debate.py",
        description="Adversarial spec debate with multiple on-node (MI355X) or remote LLMs",
        formatter_class=argparse.RawDescriptionHelpFormatter,
        epilog="""
Examples:
  echo "spec" | debate.py critique --models local/llama-3-8b
  echo "spec" | debate.py critique --models local/llama-3-8b,local/mistral-7b --focus security
  echo "spec" | debate.py critique --models stub/critique --json      # CPU plumbing
  debate.py diff --previous old.md --current new.md
  echo "spec" | debate.py export-tasks --doc-type prd
  debate.py providers | debate.py focus-areas | debate.py personas
  debate.py save-profile mine --models local/llama-3-8b --focus security

Local (MI355X) registry:
  debate.py local status
  debate.py local add-model llama-3-8b
  debate.py local alias mymodel llama-3-8b --weights /models/my-8b --gpu 2

Bedrock:
  debate.py bedrock status | enable --region us-east-1 | disable
  debate.py bedrock add-model claude-3-sonnet | alias name bedrock-id
""",
    )
    parser.add_argument("action", choices=ACTIONS, help="Action to perform")
    parser.add_argument(
        "profile_name", nargs="?",
        help="Profile name (save-profile) or bedrock/local subcommand",
    )
    # core
    parser.add_argument("--models", "-m", default=None,
                        help="Comma-separated models (e.g. local/llama-3-8b,gpt-4o)")
    parser.add_argument("--doc-type", "-d", choices=["prd", "tech"], default="tech",
                        help="Document type: prd or tech (default: tech)")
    parser.add_argument("--round", "-r", type=int, default=1, help="Current round number")
    parser.add_argument("--rounds", type=int, default=1,
                        help="Total rounds completed (used with send-final)")
    # output
    parser.add_argument("--json", "-j", action="store_true", help="Output as JSON")
    parser.add_argument("--show-cost", action="store_true",
                        help="Show cost summary after critique")
    # telegram
    parser.add_argument("--telegram", "-t", action="store_true",
                        help="Send Telegram notifications and poll for feedback")
    parser.add_argument("--poll-timeout", type=int, default=60,
                        help="Seconds to wait for Telegram reply (default: 60)")
    # critique modifiers
    parser.add_argument("--press", "-p", action="store_true",
                        help="Press models to confirm they read the full document")
    parser.add_argument("--focus", "-f",
                        help="Focus area (security, scalability, performance, ux, reliability, cost)")
    parser.add_argument("--persona", help="Persona for critique (security-engineer, ...)")
    parser.add_argument("--context", "-c", action="append", default=[],
                        help="Additional context file(s), repeatable")
    parser.add_argument("--preserve-intent", action="store_true",
                        help="Require justification for any removal or substantial change")
    # session
    parser.add_argument("--session", "-s",
                        help="Session ID for state persistence (checkpointing/resume)")
    parser.add_argument("--resume", help="Resume a previous session by ID")
    # profile
    parser.add_argument("--profile", help="Load settings from a saved profile")
    # diff
    parser.add_argument("--previous", help="Previous spec file (diff)")
    parser.add_argument("--current", help="Current spec file (diff)")
    # codex
    parser.add_argument("--codex-reasoning", default=DEFAULT_CODEX_REASONING,
                        choices=["low", "medium", "high", "xhigh"],
                        help=f"Codex CLI reasoning effort (default: {DEFAULT_CODEX_REASONING})")
    parser.add_argument("--codex-search", action="store_true",
                        help="Enable web search for Codex CLI models")
    # bedrock / local registry
    parser.add_argument("--region", help="AWS region for Bedrock")
    parser.add_argument("bedrock_arg", nargs="?",
                        help="Extra argument for bedrock/local subcommands")
    parser.add_argument("extra_arg", nargs="?",
                        help="Second extra argument (alias target)")
    parser.add_argument("--weights", help="Weights path for `local alias`")
    parser.add_argument("--gpu", type=int, default=None, help="GPU ordinal for `local alias`")
    parser.add_argument("--model-dtype", choices=["bf16", "fp8"], default=None,
                        help="Engine dtype for `local alias` (fp8 = e4m3 "
                             "MFMA prefill + fp8 weight streaming)")
    # misc
    parser.add_argument("--timeout", type=int, default=600,
                        help="Timeout in seconds for model calls (default: 600)")
    return parser


# ---------------------------------------------------------------------------
# Info / utility actions
# ---------------------------------------------------------------------------

def handle_info_command(args: argparse.Namespace) -> Optional[int]:
    """providers / focus-areas / personas / profiles / sessions listings."""
    if args.action == "providers":
        print(providers.list_providers())
        return EXIT_OK
    if args.action == "focus-areas":
        print("\n=== Focus Areas ===\n")
        for name, text in FOCUS_AREAS.items():
            first = next(
                (ln for ln in text.strip().splitlines() if ln and not ln.startswith("**")),
                "",
            )
            print(f"  {name:<12} {first}")
        return EXIT_OK
    if args.action == "personas":
        print("\n=== Personas ===\n")
        for name, text in PERSONAS.items():
            print(f"  {name:<26} {text[:90]}...")
        return EXIT_OK
    if args.action == "profiles":
        names = providers.list_profiles()
        if not names:
            print("No saved profiles.")
        else:
            print("\n=== Saved Profiles ===\n")
            for n in names:
                data = providers.load_profile(n) or {}
                print(f"  {n}: models={data.get('models')}, focus={data.get('focus')}")
        return EXIT_OK
    if args.action == "sessions":
        sessions = SessionState.list_sessions()
        if not sessions:
            print("No saved sessions.")
        else:
            print("\n=== Saved Sessions ===\n")
            for s in sessions:
                print(
                    f"  {s['session_id']}: round {s['round']}, {s['doc_type']}, "
                    f"models={','.join(s['models'])}, updated {s['updated_at']}"
                )
        return EXIT_OK
    return None


def handle_utility_command(args: argparse.Namespace) -> Optional[int]:
    """diff / save-profile / bedrock / local."""
    if args.action == "diff":
        if not args.previous or not args.current:
            print("Error: diff requires --previous and --current", file=sys.stderr)
            return EXIT_ERROR
        try:
            from pathlib import Path

            prev = Path(args.previous).read_text()
            curr = Path(args.current).read_text()
        except OSError as e:
            print(f"Error: {e}", file=sys.stderr)
            return EXIT_ERROR
        diff = generate_diff(prev, curr)
        print(diff if diff else "No differences found.")
        return EXIT_OK
    if args.action == "save-profile":
        if not args.profile_name:
            print("Error: save-profile requires a profile name", file=sys.stderr)
            return EXIT_ERROR
        settings = {
            "models": args.models,
            "doc_type": args.doc_type,
            "focus": args.focus,
            "persona": args.persona,
            "context": args.context,
            "preserve_intent": args.preserve_intent,
        }
        path = providers.save_profile(args.profile_name, settings)
        print(f"Profile saved: {path}")
        return EXIT_OK
    if args.action == "bedrock":
        return providers.handle_bedrock_command(
            args.profile_name, args.bedrock_arg, args.extra_arg, args.region
        )
    if args.action == "local":
        return providers.handle_local_command(
            args.profile_name, args.bedrock_arg, args.extra_arg,
            args.weights, args.gpu, args.model_dtype,
        )
    return None


def apply_profile(args: argparse.Namespace) -> None:
    """Fill only unset flags from --profile (explicit flags win;
    reference: debate.py:529-550)."""
    if not args.profile:
        return
    data = providers.load_profile(args.profile)
    if data is None:
        print(f"Error: profile not found: {args.profile}", file=sys.stderr)
        sys.exit(EXIT_ERROR)
    if args.models is None and data.get("models"):
        args.models = data["models"] if isinstance(data["models"], str) else ",".join(data["models"])
    if args.doc_type == "tech" and data.get("doc_type"):
        args.doc_type = data["doc_type"]
    if not args.focus and data.get("focus"):
        args.focus = data["focus"]
    if not args.persona and data.get("persona"):
        args.persona = data["persona"]
    if not args.context and data.get("context"):
        args.context = data["context"]
    if not args.preserve_intent and data.get("preserve_intent"):
        args.preserve_intent = data["preserve_intent"]


def parse_models(args: argparse.Namespace) -> list[str]:
    if args.models:
        return [m.strip() for m in args.models.split(",") if m.strip()]
    if args.resume:
        return []  # models come from the resumed session
    default = providers.get_default_model()
    if default is None:
        print(
            "Error: no models specified and no provider configured.\n"
            "Set an API key, register a local model, or pass --models.",
            file=sys.stderr,
        )
        sys.exit(EXIT_CONFIG)
    print(f"Using default model: {default}", file=sys.stderr)
    return [default]


def setup_bedrock(args: argparse.Namespace, models: list[str]) -> tuple[bool, Optional[str]]:
    """(bedrock_mode, region); exit 2 when bedrock models are not enabled
    (reference: debate.py:614-667)."""
    config = providers.get_bedrock_config()
    if not config.get("enabled"):
        return False, None
    valid, invalid = providers.validate_bedrock_models(models, config)
    if invalid:
        print(
            f"Error: model(s) not in Bedrock available list: {', '.join(invalid)}\n"
            "Add them with: debate.py bedrock add-model <name>",
            file=sys.stderr,
        )
        sys.exit(EXIT_CONFIG)
    return True, args.region or config.get("region")


def validate_models_before_run(models: list[str], bedrock_mode: bool) -> None:
    """Fail fast (exit 2) when credentials/engines are missing
    (reference: debate.py:976-1022)."""
    if bedrock_mode:
        return
    import os

    if os.environ.get("ADVSPEC_BACKEND") == "stub":
        return
    to_check = [m for m in models if not m.startswith("stub")]
    valid, invalid = providers.validate_model_credentials(to_check)
    if invalid:
        print(
            f"Error: missing credentials or unavailable backend for: {', '.join(invalid)}",
            file=sys.stderr,
        )
        sys.exit(EXIT_CONFIG)


# ---------------------------------------------------------------------------
# send-final / export-tasks
# ---------------------------------------------------------------------------

def handle_send_final(args: argparse.Namespace, models: list[str]) -> int:
    spec = sys.stdin.read().strip()
    if not spec:
        print("Error: No spec provided via stdin", file=sys.stderr)
        return EXIT_ERROR
    token, chat_id = telegram.get_config()
    if not token or not chat_id:
        print("Error: TELEGRAM_BOT_TOKEN / TELEGRAM_CHAT_ID not set", file=sys.stderr)
        return EXIT_CONFIG
    doc_name = get_doc_type_name(args.doc_type)
    header = (
        f"FINAL {doc_name}\n"
        f"Rounds: {args.rounds} | Models: {', '.join(models)}\n"
        + "=" * 30 + "\n"
    )
    if telegram.send_long_message(token, chat_id, header + spec):
        print("Final document sent to Telegram.")
        return EXIT_OK
    print("Failed to send final document to Telegram.", file=sys.stderr)
    return EXIT_ERROR


def handle_export_tasks(args: argparse.Namespace, models: list[str]) -> int:
    spec = sys.stdin.read().strip()
    if not spec:
        print("Error: No spec provided via stdin", file=sys.stderr)
        return EXIT_ERROR
    doc_type_name = get_doc_type_name(args.doc_type)
    prompt = EXPORT_TASKS_PROMPT.format(doc_type_name=doc_type_name, spec=spec)
    from ..engine.backend import get_backend, is_o_series_model

    backend = get_backend(models[0], codex_reasoning=args.codex_reasoning)
    try:
        # export-tasks samples at temperature 0.3 (reference: debate.py:713),
        # except o-series models which reject custom temperature.
        temp = 0.3 if not is_o_series_model(models[0]) else 1.0
        content, _, _ = backend.generate(
            "", prompt, max_tokens=8000, temperature=temp, timeout=args.timeout
        )
        tasks = extract_tasks(content)
    except Exception as e:
        print(f"Error: {e}", file=sys.stderr)
        return EXIT_ERROR
    if args.json:
        print(json.dumps({"tasks": tasks}, indent=2))
    else:
        print(f"\n=== Extracted {len(tasks)} Tasks ===\n")
        for i, task in enumerate(tasks, 1):
            print(
                f"{i}. [{task.get('type', 'task')}] [{task.get('priority', 'medium')}] "
                f"{task.get('title', 'Untitled')}"
            )
            if task.get("description"):
                print(f"   {task['description'][:100]}...")
            if task.get("acceptance_criteria"):
                print(f"   Acceptance criteria: {len(task['acceptance_criteria'])} items")
            print()
    return EXIT_OK


# ---------------------------------------------------------------------------
# critique
# ---------------------------------------------------------------------------

def load_or_resume_session(
    args: argparse.Namespace, models: list[str]
) -> tuple[str, Optional[SessionState], list[str]]:
    session_state = None
    if args.resume:
        try:
            session_state = SessionState.load(args.resume)
        except FileNotFoundError as e:
            print(f"Error: {e}", file=sys.stderr)
            sys.exit(EXIT_CONFIG)
        print(
            f"Resuming session '{args.resume}' at round {session_state.round}",
            file=sys.stderr,
        )
        spec = session_state.spec
        args.round = session_state.round
        args.doc_type = session_state.doc_type
        args.models = ",".join(session_state.models)
        if session_state.focus:
            args.focus = session_state.focus
        if session_state.persona:
            args.persona = session_state.persona
        if session_state.preserve_intent:
            args.preserve_intent = session_state.preserve_intent
        models = session_state.models
    else:
        spec = sys.stdin.read().strip()
        if not spec:
            print("Error: No spec provided via stdin", file=sys.stderr)
            sys.exit(EXIT_ERROR)
    if args.session and not session_state:
        from datetime import datetime

        session_state = SessionState(
            session_id=args.session,
            spec=spec,
            round=args.round,
            doc_type=args.doc_type,
            models=models,
            focus=args.focus,
            persona=args.persona,
            preserve_intent=args.preserve_intent,
            created_at=datetime.now().isoformat(),
        )
        session_state.save()
        print(f"Session '{args.session}' created", file=sys.stderr)
    return spec, session_state, models


def send_telegram_notification(
    models: list[str], round_num: int, results: list[ModelResponse], poll_timeout: int
) -> Optional[str]:
    """Round summary to Telegram; long-poll for human feedback
    (reference: debate.py:96-169)."""
    token, chat_id = telegram.get_config()
    if not token or not chat_id:
        print("Telegram not configured; skipping notification.", file=sys.stderr)
        return None
    lines = [f"Round {round_num} results:"]
    for r in results:
        if r.error:
            lines.append(f"  {r.model}: ERROR - {r.error[:100]}")
        elif r.agreed:
            lines.append(f"  {r.model}: AGREE")
        else:
            lines.append(f"  {r.model}: critique ({protocol.get_critique_summary(r.response, 120)})")
    successful = [r for r in results if not r.error]
    if successful and all(r.agreed for r in successful):
        lines.append("ALL MODELS AGREE")
    lines.append(f"Cost so far: ${cost_tracker.total_cost:.4f}")
    lines.append("")
    lines.append(f"Reply within {poll_timeout}s to inject feedback into the debate.")
    watermark = telegram.get_last_update_id(token)
    if not telegram.send_long_message(token, chat_id, "\n".join(lines)):
        return None
    return telegram.poll_for_reply(token, chat_id, watermark, timeout=poll_timeout)


def run_critique(
    args: argparse.Namespace,
    spec: str,
    models: list[str],
    session_state: Optional[SessionState],
    context: Optional[str],
    bedrock_mode: bool,
    bedrock_region: Optional[str],
) -> int:
    mode = "pressing for confirmation" if args.press else "critiquing"
    extras = "".join(
        [
            f" (focus: {args.focus})" if args.focus else "",
            f" (persona: {args.persona})" if args.persona else "",
            " (preserve-intent)" if args.preserve_intent else "",
            " (search)" if args.codex_search else "",
        ]
    )
    print(
        f"Calling {len(models)} model(s) ({mode}){extras}: {', '.join(models)}...",
        file=sys.stderr,
    )

    results = call_models_parallel(
        models,
        spec,
        args.round,
        args.doc_type,
        args.press,
        args.focus,
        args.persona,
        context,
        args.preserve_intent,
        args.codex_reasoning,
        args.codex_search,
        args.timeout,
        bedrock_mode,
        bedrock_region,
    )

    for r in results:
        if r.error:
            print(f"Warning: {r.model} returned error: {r.error}", file=sys.stderr)

    successful = [r for r in results if not r.error]
    all_agreed = all(r.agreed for r in successful) if successful else False

    session_id = session_state.session_id if session_state else args.session
    if session_id or args.session:
        save_checkpoint(spec, args.round, session_id)

    # Deterministic "latest spec": first spec in MODEL-LIST order (results
    # are already ordered; the reference used nondeterministic arrival
    # order, debate.py:859-863).
    latest_spec = spec
    for r in successful:
        if r.spec:
            latest_spec = r.spec
            break

    if session_state:
        session_state.spec = latest_spec
        session_state.round = args.round + 1
        session_state.history.append(
            {
                "round": args.round,
                "all_agreed": all_agreed,
                "models": [
                    {"model": r.model, "agreed": r.agreed, "error": r.error}
                    for r in results
                ],
            }
        )
        session_state.save()

    user_feedback = None
    if args.telegram:
        user_feedback = send_telegram_notification(
            models, args.round, results, args.poll_timeout
        )
        if user_feedback:
            print(f"Received feedback: {user_feedback}", file=sys.stderr)

    output_results(args, results, models, all_agreed, user_feedback, session_state)
    return EXIT_OK


def output_results(
    args: argparse.Namespace,
    results: list[ModelResponse],
    models: list[str],
    all_agreed: bool,
    user_feedback: Optional[str],
    session_state: Optional[SessionState],
) -> None:
    """JSON or text round output (schema: reference debate.py:909-941)."""
    if args.json:
        output: dict[str, Any] = {
            "all_agreed": all_agreed,
            "round": args.round,
            "doc_type": args.doc_type,
            "models": models,
            "focus": args.focus,
            "persona": args.persona,
            "preserve_intent": args.preserve_intent,
            "session": session_state.session_id if session_state else args.session,
            "results": [
                {
                    "model": r.model,
                    "agreed": r.agreed,
                    "response": r.response,
                    "spec": r.spec,
                    "error": r.error,
                    "input_tokens": r.input_tokens,
                    "output_tokens": r.output_tokens,
                    "cost": r.cost,
                }
                for r in results
            ],
            "cost": {
                "total": cost_tracker.total_cost,
                "input_tokens": cost_tracker.total_input_tokens,
                "output_tokens": cost_tracker.total_output_tokens,
                "by_model": cost_tracker.by_model,
            },
        }
        if user_feedback:
            output["user_feedback"] = user_feedback
        print(json.dumps(output, indent=2))
    else:
        doc_type_name = get_doc_type_name(args.doc_type)
        print(f"\n=== Round {args.round} Results ({doc_type_name}) ===\n")
        for r in results:
            print(f"--- {r.model} ---")
            if r.error:
                print(f"ERROR: {r.error}")
            elif r.agreed:
                print("[AGREE]")
            else:
                print(r.response)
            print()
        if all_agreed:
            print("=== ALL MODELS AGREE ===")
        else:
            successful = [r for r in results if not r.error]
            agreed = [r.model for r in successful if r.agreed]
            disagreed = [r.model for r in successful if not r.agreed]
            if agreed:
                print(f"Agreed: {', '.join(agreed)}")
            if disagreed:
                print(f"Critiqued: {', '.join(disagreed)}")
        if user_feedback:
            print()
            print("=== User Feedback ===")
            print(user_feedback)
        if args.show_cost:
            print(cost_tracker.summary())


# ---------------------------------------------------------------------------
# main
# ---------------------------------------------------------------------------

def handle_serve(args: argparse.Namespace) -> int:
    """`serve [status|stop]`: persistent engine daemon (daemon.py).

    Round 2+ of a skill session skips the ~40 s per-process model init:
    `critique` transparently forwards to a live daemon whose engine cache
    keeps opponent weights resident in HBM3E between rounds.
    """
    from .. import daemon

    sub = args.profile_name
    if sub == "status":
        alive = daemon.ping()
        print(f"daemon: {'running' if alive else 'not running'} "
              f"({daemon.SOCKET_PATH})")
        return EXIT_OK if alive else EXIT_ERROR
    if sub == "stop":
        ok = daemon.stop()
        print("daemon: stopped" if ok else "daemon: not running")
        return EXIT_OK
    if sub not in (None, "start"):
        print(f"Unknown serve subcommand: {sub}", file=sys.stderr)
        return EXIT_ERROR
    srv = daemon.serve()
    print(f"adversarial-spec daemon listening on {daemon.SOCKET_PATH}",
          file=sys.stderr)
    try:
        srv.serve_forever()
    except KeyboardInterrupt:
        pass
    finally:
        srv.server_close()
    return EXIT_OK


def main(argv: Optional[list[str]] = None) -> int:
    parser = create_parser()
    args = parser.parse_args(argv)

    if args.action == "serve":
        return handle_serve(args)

    # a live daemon serves the expensive actions with warm engines; same
    # argv + stdin forwarded, stdout/exit relayed (ADVSPEC_NO_DAEMON=1 or
    # no daemon -> unchanged in-process path)
    if args.action in ("critique", "export-tasks"):
        from .. import daemon

        raw_argv = list(argv) if argv is not None else sys.argv[1:]
        if daemon.ping():
            stdin_text = "" if args.resume else sys.stdin.read()
            fwd = daemon.try_forward(raw_argv, stdin_text)
            if fwd is not None:
                code, out, err = fwd
                sys.stdout.write(out)
                sys.stderr.write(err)
                return code
            # daemon vanished mid-flight: fall through with the spec we read
            if not args.resume:
                sys.stdin = __import__("io").StringIO(stdin_text)

    code = handle_info_command(args)
    if code is not None:
        return code
    code = handle_utility_command(args)
    if code is not None:
        return code

    apply_profile(args)
    models = parse_models(args)
    context = load_context_files(args.context)
    bedrock_mode, bedrock_region = setup_bedrock(args, models)

    if args.action == "send-final":
        return handle_send_final(args, models)

    if args.action == "export-tasks":
        validate_models_before_run(models, bedrock_mode)
        return handle_export_tasks(args, models)

    # critique: a resumed session may supply the model list, so validation
    # runs after session load.
    spec, session_state, models = load_or_resume_session(args, models)
    validate_models_before_run(models, bedrock_mode)
    try:
        return run_critique(
            args, spec, models, session_state, context, bedrock_mode, bedrock_region
        )
    except Exception as e:
        print(f"Error: {e}", file=sys.stderr)
        return EXIT_ERROR


if __name__ == "__main__":
    sys.exit(main())
