"""Engine daemon round-trip (debate.py serve): warm-process forwarding."""

from __future__ import annotations

import json
import threading

import pytest

from adversarial_spec_amd import daemon
from adversarial_spec_amd.cli import debate as cli

SPEC = "# Payments API\n\nA spec that needs critique."


@pytest.fixture
def live_daemon(tmp_path, monkeypatch):
    sock = tmp_path / "d.sock"
    monkeypatch.setattr(daemon, "SOCKET_PATH", sock)
    srv = daemon.serve(sock)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield sock, srv
    srv.shutdown()
    srv.server_close()


class TestDaemon:
    def test_ping(self, live_daemon):
        sock, _srv = live_daemon
        assert daemon.ping(sock)
        assert not daemon.ping(sock.parent / "missing.sock")

    def test_forward_critique_round_trip(self, live_daemon, monkeypatch):
        sock, srv = live_daemon
        monkeypatch.delenv("ADVSPEC_IN_DAEMON", raising=False)
        argv = ["critique", "--models", "stub/agree", "--json"]
        for i in range(2):  # second request hits the WARM daemon process
            fwd = daemon.try_forward(argv, SPEC, sock)
            assert fwd is not None, "daemon did not answer"
            code, out, err = fwd
            assert code == 0, err
            payload = json.loads(out)
            assert payload["all_agreed"] is True
            assert payload["results"][0]["model"] == "stub/agree"
        assert srv.requests_served == 2

    def test_no_forward_inside_daemon(self, live_daemon, monkeypatch):
        sock, _ = live_daemon
        monkeypatch.setenv("ADVSPEC_IN_DAEMON", "1")
        assert daemon.try_forward(["critique"], SPEC, sock) is None
        monkeypatch.delenv("ADVSPEC_IN_DAEMON")
        monkeypatch.setenv("ADVSPEC_NO_DAEMON", "1")
        assert daemon.try_forward(["critique"], SPEC, sock) is None

    def test_cli_serve_status_and_stop(self, live_daemon, monkeypatch, capsys):
        sock, srv = live_daemon
        assert cli.main(["serve", "status"]) == 0
        assert "running" in capsys.readouterr().out
        assert daemon.stop(sock)
        # server thread shuts down; give it a beat
        import time

        for _ in range(50):
            if not daemon.ping(sock):
                break
            time.sleep(0.05)

    def test_cli_critique_uses_daemon(self, live_daemon, monkeypatch, capsys):
        """debate.py critique transparently forwards when a daemon runs."""
        sock, srv = live_daemon
        monkeypatch.delenv("ADVSPEC_IN_DAEMON", raising=False)
        monkeypatch.delenv("ADVSPEC_NO_DAEMON", raising=False)
        import io
        import sys as _sys

        monkeypatch.setattr(_sys, "stdin", io.StringIO(SPEC))
        code = cli.main(["critique", "--models", "stub/agree", "--json"])
        assert code == 0
        payload = json.loads(capsys.readouterr().out)
        assert payload["all_agreed"] is True
        assert srv.requests_served == 1


class TestDaemonGuards:
    def test_second_serve_refused_while_live(self, live_daemon):
        sock, _srv = live_daemon
        with pytest.raises(RuntimeError, match="already serving"):
            daemon.serve(sock)

    def test_stale_socket_cleaned_up(self, tmp_path):
        sock = tmp_path / "stale.sock"
        sock.touch()  # dead socket file, nothing listening
        srv = daemon.serve(sock)
        try:
            assert sock.exists()
        finally:
            srv.server_close()

    def test_serve_creates_deep_socket_dir(self, tmp_path):
        """serve() must mkdir the socket's parents (mutation killer:
        mkdir(parents=True) — a fresh install has no config dir)."""
        sock = tmp_path / "a" / "b" / "daemon.sock"
        srv = daemon.serve(sock)
        try:
            assert sock.exists()
        finally:
            srv.server_close()

    def test_dispatch_error_reports_code_1(self, live_daemon, monkeypatch):
        """Both daemon error paths answer with exit code EXACTLY 1 (the
        reference's processing-error code; 2 means missing credentials)."""
        sock, _srv = live_daemon
        from adversarial_spec_amd.cli import debate as cli_mod

        def boom(argv):
            raise RuntimeError("engine exploded")

        # inner path: cli.main raises -> caught inside run_request
        monkeypatch.setattr(cli_mod, "main", boom)
        fwd = daemon.try_forward(["critique"], SPEC, sock)
        assert fwd is not None and fwd[0] == 1
        # dispatch path: run_request itself raises -> handler catches
        monkeypatch.setattr(daemon, "run_request",
                            lambda argv, s: (_ for _ in ()).throw(
                                RuntimeError("dispatch")))
        fwd = daemon.try_forward(["critique"], SPEC, sock)
        assert fwd is not None and fwd[0] == 1
        assert "dispatch error" in fwd[2]

    def test_concurrent_forwards(self, live_daemon):
        """Two critiques in flight at once (the threaded server + request
        lock must not deadlock or cross wires). Uses the raw socket
        request: with an IN-PROCESS server the ADVSPEC_IN_DAEMON guard is
        briefly visible to the client threads too (same environment — a
        real daemon is a separate process), so try_forward could
        legitimately decline mid-race."""
        import concurrent.futures as cf

        sock, srv = live_daemon
        payload = {"argv": ["critique", "--models", "stub/agree", "--json"],
                   "stdin": SPEC}
        with cf.ThreadPoolExecutor(2) as pool:
            futs = [pool.submit(daemon._request, payload, sock)
                    for _ in range(2)]
            results = [f.result(timeout=60) for f in futs]
        for r in results:
            assert r is not None and r["code"] == 0
        assert srv.requests_served == 2

    def test_cli_serve_stop_subcommand(self, live_daemon, capsys):
        """`debate.py serve stop` must hit the stop dispatch (exit 0 and
        the daemon actually goes down)."""
        sock, _srv = live_daemon
        assert cli.main(["serve", "stop"]) == 0
        assert "stopped" in capsys.readouterr().out
        import time as _t

        for _ in range(50):
            if not daemon.ping(sock):
                break
            _t.sleep(0.05)
        assert not daemon.ping(sock)
