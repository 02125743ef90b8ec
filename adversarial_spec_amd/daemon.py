"""Persistent engine daemon: `debate.py serve`.

The CLI is one process per debate round (reference UX, SKILL.md steps),
so every round re-pays model init (~40 s random-init / ~15 s safetensors
for an 8B opponent). The daemon keeps the process-wide engine cache
(engine/local.py get_engine: weights resident in HBM3E) alive across
rounds behind a local unix socket:

  debate.py serve            # foreground server (skill backgrounds it)
  debate.py serve status     # is a daemon listening?
  debate.py serve stop       # shut it down

`debate.py critique` transparently forwards to a running daemon (same
argv + stdin spec, stdout/exit code relayed), so round 2+ of a skill
session runs at warm-engine speed with an unchanged single-invocation
surface. No daemon (or ADVSPEC_NO_DAEMON=1) -> the CLI runs in-process
exactly as before.

Wire format: one JSON line per request/response over the socket —
  request:  {"argv": [...], "stdin": "<spec>"}
  control:  {"op": "ping"} / {"op": "stop"}
  response: {"code": int, "stdout": str, "stderr": str}
"""

from __future__ import annotations

import contextlib
import io
import json
import os
import socket
import socketserver
import sys
import threading
from pathlib import Path
from typing import Optional

SOCKET_PATH = Path(
    os.environ.get(
        "ADVSPEC_DAEMON_SOCKET",
        str(Path.home() / ".config" / "adversarial-spec" / "daemon.sock"),
    )
)

# stop forwarding loops: set in the daemon process before re-entering main()
_IN_DAEMON_ENV = "ADVSPEC_IN_DAEMON"

# one CLI invocation at a time: run_request redirects the PROCESS-global
# stdio and environment, so concurrent requests must queue (they would
# interleave output otherwise — caught by the concurrency test). The
# expensive inner work is engine inference, which the per-engine generate
# locks serialize per opponent anyway.
_REQUEST_LOCK = threading.Lock()


def _recv_line(sock: socket.socket, limit: int = 64 * 1024 * 1024) -> bytes:
    chunks = []
    n = 0
    while n < limit:
        b = sock.recv(65536)
        if not b:
            break
        chunks.append(b)
        n += len(b)
        if b.endswith(b"\n"):
            break
    return b"".join(chunks)


class _Handler(socketserver.StreamRequestHandler):
    def handle(self) -> None:  # pragma: no cover - exercised via round-trip
        try:
            line = self.rfile.readline(64 * 1024 * 1024)
            req = json.loads(line.decode("utf-8"))
        except Exception:
            return
        srv: "_Server" = self.server  # type: ignore[assignment]
        if req.get("op") == "ping":
            self.wfile.write(b'{"ok": true}\n')
            return
        if req.get("op") == "stop":
            self.wfile.write(b'{"ok": true}\n')
            threading.Thread(target=srv.shutdown, daemon=True).start()
            return
        try:
            code, out, err = run_request(req.get("argv", []),
                                         req.get("stdin", ""))
        except Exception as e:  # noqa: BLE001 - the daemon must answer
            code, out, err = 1, "", f"daemon dispatch error: {e!r}"
        srv.requests_served += 1
        resp = json.dumps({"code": code, "stdout": out, "stderr": err})
        self.wfile.write(resp.encode("utf-8") + b"\n")


class _Server(socketserver.ThreadingUnixStreamServer):
    daemon_threads = True
    allow_reuse_address = True
    requests_served = 0


def run_request(argv: list, stdin_text: str) -> tuple[int, str, str]:
    """Execute one CLI invocation inside the daemon process (serialized:
    stdio/env redirection is process-global)."""
    from .cli import debate as cli

    with _REQUEST_LOCK:
        return _run_request_locked(cli, argv, stdin_text)


def _run_request_locked(cli, argv: list, stdin_text: str) -> tuple[int, str, str]:
    prev_env = os.environ.get(_IN_DAEMON_ENV)
    os.environ[_IN_DAEMON_ENV] = "1"
    out_buf, err_buf = io.StringIO(), io.StringIO()
    old_stdin = sys.stdin
    try:
        sys.stdin = io.StringIO(stdin_text)
        with contextlib.redirect_stdout(out_buf), \
                contextlib.redirect_stderr(err_buf):
            try:
                code = cli.main(list(argv))
            except SystemExit as e:  # argparse errors etc.
                code = int(e.code or 0)
            except Exception as e:  # noqa: BLE001 - daemon must not die
                print(f"daemon error: {e}", file=sys.stderr)
                code = 1
    finally:
        sys.stdin = old_stdin
        # restore rather than leave set: threaded in-process servers (the
        # test fixture) share this environment with their clients
        if prev_env is None:
            os.environ.pop(_IN_DAEMON_ENV, None)
        else:
            os.environ[_IN_DAEMON_ENV] = prev_env
    return code, out_buf.getvalue(), err_buf.getvalue()


def serve(socket_path: Optional[Path] = None) -> _Server:
    """Bind the unix socket and return the server (caller serves forever).

    Refuses to displace a LIVE daemon (a second `serve` would silently
    steal the socket and strand the first process's warm engines); a
    stale socket file from a dead daemon is cleaned up.
    """
    path = Path(socket_path or SOCKET_PATH)
    if ping(path):
        raise RuntimeError(
            f"an adversarial-spec daemon is already serving on {path} "
            "(stop it with: debate.py serve stop)"
        )
    path.parent.mkdir(parents=True, exist_ok=True)
    with contextlib.suppress(FileNotFoundError):
        path.unlink()
    srv = _Server(str(path), _Handler)
    return srv


def _request(payload: dict, socket_path: Optional[Path] = None,
             timeout: float = 3600.0) -> Optional[dict]:
    path = Path(socket_path or SOCKET_PATH)
    if not path.exists():
        return None
    try:
        with socket.socket(socket.AF_UNIX, socket.SOCK_STREAM) as s:
            s.settimeout(timeout)
            s.connect(str(path))
            s.sendall(json.dumps(payload).encode("utf-8") + b"\n")
            data = _recv_line(s)
        return json.loads(data.decode("utf-8")) if data else None
    except (OSError, json.JSONDecodeError):
        return None


def ping(socket_path: Optional[Path] = None) -> bool:
    r = _request({"op": "ping"}, socket_path, timeout=5.0)
    return bool(r and r.get("ok"))


def stop(socket_path: Optional[Path] = None) -> bool:
    r = _request({"op": "stop"}, socket_path, timeout=10.0)
    return bool(r and r.get("ok"))


def try_forward(argv: list, stdin_text: str,
                socket_path: Optional[Path] = None) -> Optional[tuple[int, str, str]]:
    """Forward a CLI invocation to a live daemon; None -> run locally."""
    if os.environ.get(_IN_DAEMON_ENV) or os.environ.get("ADVSPEC_NO_DAEMON"):
        return None
    r = _request({"argv": list(argv), "stdin": stdin_text}, socket_path)
    if r is None or "code" not in r:
        return None
    return int(r["code"]), r.get("stdout", ""), r.get("stderr", "")
