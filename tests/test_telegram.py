"""Telegram channel tests: chunking, API wrapper, polling (all mocked)."""

import json
from unittest.mock import MagicMock, patch

import pytest

from adversarial_spec_amd import telegram


class TestSplitMessage:
    def test_short_untouched(self):
        assert telegram.split_message("hi") == ["hi"]

    def test_exact_limit(self):
        text = "x" * telegram.MAX_MESSAGE_LENGTH
        assert telegram.split_message(text) == [text]

    def test_over_limit_splits(self):
        text = "x" * (telegram.MAX_MESSAGE_LENGTH + 1)
        chunks = telegram.split_message(text)
        assert len(chunks) == 2
        assert all(len(c) <= telegram.MAX_MESSAGE_LENGTH for c in chunks)

    def test_prefers_newline(self):
        text = "a" * 4000 + "\n" + "b" * 1000
        chunks = telegram.split_message(text)
        assert chunks[0] == "a" * 4000
        assert chunks[1] == "b" * 1000

    def test_reassembles(self):
        text = ("line\n" * 3000).strip()
        chunks = telegram.split_message(text)
        assert "".join(c.replace("\n", "") for c in chunks) == text.replace("\n", "")


def _response(payload: dict) -> MagicMock:
    m = MagicMock()
    m.read.return_value = json.dumps(payload).encode()
    m.__enter__ = lambda s: m
    m.__exit__ = MagicMock(return_value=False)
    return m


class TestApiCall:
    def test_ok(self):
        with patch.object(telegram.urllib.request, "urlopen",
                          return_value=_response({"ok": True, "result": []})):
            out = telegram.api_call("tok", "getUpdates")
            assert out["ok"] is True

    def test_api_not_ok_raises(self):
        with patch.object(telegram.urllib.request, "urlopen",
                          return_value=_response({"ok": False, "description": "bad"})):
            with pytest.raises(RuntimeError, match="bad"):
                telegram.api_call("tok", "sendMessage")

    def test_http_error_raises(self):
        import urllib.error

        def boom(*a, **k):
            raise urllib.error.HTTPError("u", 403, "forbidden", {}, None)

        with patch.object(telegram.urllib.request, "urlopen", side_effect=boom):
            with pytest.raises(RuntimeError, match="403"):
                telegram.api_call("tok", "sendMessage")

    def test_url_error_raises(self):
        import urllib.error

        with patch.object(telegram.urllib.request, "urlopen",
                          side_effect=urllib.error.URLError("down")):
            with pytest.raises(RuntimeError, match="unreachable"):
                telegram.api_call("tok", "sendMessage")


class TestSend:
    def test_send_message_ok(self):
        with patch.object(telegram, "api_call", return_value={"ok": True}):
            assert telegram.send_message("t", "c", "hello")

    def test_send_message_fail(self, capsys):
        with patch.object(telegram, "api_call", side_effect=RuntimeError("x")):
            assert not telegram.send_message("t", "c", "hello")
        capsys.readouterr()

    def test_send_long_chunks_with_sleep(self):
        sent = []
        with patch.object(telegram, "send_message",
                          side_effect=lambda t, c, m: sent.append(m) or True), \
             patch.object(telegram.time, "sleep") as sl:
            text = "x" * (telegram.MAX_MESSAGE_LENGTH * 2 + 10)
            assert telegram.send_long_message("t", "c", text)
        assert len(sent) == 3
        assert sl.call_count == 2
        sl.assert_called_with(telegram.CHUNK_SLEEP)


class TestPoll:
    def test_get_last_update_id(self):
        with patch.object(telegram, "api_call",
                          return_value={"ok": True, "result": [{"update_id": 7}, {"update_id": 9}]}):
            assert telegram.get_last_update_id("t") == 9

    def test_poll_finds_reply(self):
        payload = {
            "ok": True,
            "result": [
                {"update_id": 10,
                 "message": {"chat": {"id": 42}, "text": "looks good"}},
            ],
        }
        with patch.object(telegram, "api_call", return_value=payload):
            reply = telegram.poll_for_reply("t", "42", 9, timeout=5)
        assert reply == "looks good"

    def test_poll_filters_other_chat(self):
        payload = {
            "ok": True,
            "result": [
                {"update_id": 10, "message": {"chat": {"id": 99}, "text": "noise"}},
            ],
        }
        clock = {"t": 0}

        def fake_time():
            clock["t"] += 1
            return clock["t"]

        with patch.object(telegram, "api_call", return_value=payload), \
             patch.object(telegram.time, "time", side_effect=fake_time):
            reply = telegram.poll_for_reply("t", "42", 9, timeout=5)
        assert reply is None

    def test_poll_timeout_none(self):
        clock = {"t": 0}

        def fake_time():
            clock["t"] += 3
            return clock["t"]

        with patch.object(telegram, "api_call", return_value={"ok": True, "result": []}), \
             patch.object(telegram.time, "time", side_effect=fake_time):
            assert telegram.poll_for_reply("t", "42", 0, timeout=5) is None


class TestCLI:
    def test_no_token_exits_2(self, clean_env, capsys):
        assert telegram.main(["send", "--message", "x"]) == 2
        capsys.readouterr()

    def test_send(self, clean_env, capsys):
        clean_env.setenv("TELEGRAM_BOT_TOKEN", "t")
        clean_env.setenv("TELEGRAM_CHAT_ID", "c")
        with patch.object(telegram, "send_long_message", return_value=True):
            assert telegram.main(["send", "--message", "hello"]) == 0
        capsys.readouterr()


class TestMutationKillers:
    """Pin behaviors the example tests left unobserved (mutation-driven)."""

    def test_poll_offset_semantics(self):
        """getUpdates must start EXACTLY after the watermark (off-by-one
        re-reads or skips an update) and ack past the consumed update."""
        calls = []

        def fake_api(token, method, params=None, timeout=35):
            calls.append(dict(params or {}))
            return {"ok": True, "result": [
                {"update_id": 10,
                 "message": {"chat": {"id": 42}, "text": "ok"}}]}

        with patch.object(telegram, "api_call", side_effect=fake_api):
            assert telegram.poll_for_reply("t", "42", 9, timeout=5) == "ok"
        assert calls[0]["offset"] == 10   # watermark + 1, exactly
        assert calls[-1]["offset"] == 11  # ack = consumed update_id + 1

    def test_discover_offset_semantics(self):
        calls = []

        def fake_api(token, method, params=None, timeout=35):
            calls.append(dict(params or {}))
            if len(calls) == 1:  # get_last_update_id probe
                return {"ok": True, "result": [{"update_id": 4}]}
            return {"ok": True, "result": [
                {"update_id": 5, "message": {"chat": {"id": 7}}}]}

        with patch.object(telegram, "api_call", side_effect=fake_api):
            assert telegram.discover_chat_id("t", wait=5) == "7"
        assert calls[1]["offset"] == 5  # last_update_id + 1, exactly

    def test_send_long_failure_propagates(self):
        """A failed chunk must fail the WHOLE send (silent-success bug)."""
        with patch.object(telegram, "send_message", return_value=False):
            assert telegram.send_long_message("t", "c", "hello") is False

    def test_cli_exit_codes_exact(self, clean_env, capsys):
        clean_env.setenv("TELEGRAM_BOT_TOKEN", "t")
        # setup finds nothing -> 1
        with patch.object(telegram, "discover_chat_id", return_value=None):
            assert telegram.main(["setup", "--timeout", "1"]) == 1
        # chat id missing for send -> 2
        assert telegram.main(["send", "--message", "x"]) == 2
        clean_env.setenv("TELEGRAM_CHAT_ID", "c")
        # send failure -> 1
        with patch.object(telegram, "send_long_message", return_value=False):
            assert telegram.main(["send", "--message", "x"]) == 1
        # poll dispatch: reply -> 0 and prints it; none -> 1
        with patch.object(telegram, "get_last_update_id", return_value=0), \
             patch.object(telegram, "poll_for_reply", return_value="yes"):
            assert telegram.main(["poll"]) == 0
        assert "yes" in capsys.readouterr().out
        with patch.object(telegram, "get_last_update_id", return_value=0), \
             patch.object(telegram, "poll_for_reply", return_value=None):
            assert telegram.main(["poll"]) == 1
        capsys.readouterr()

    def test_split_no_empty_chunk_on_leading_newline(self):
        """rfind can return 0 (newline at position 0): the cut<=0 guard
        must treat it as no-boundary, never emitting an empty chunk."""
        chunks = telegram.split_message("\n" + "x" * 50, 10)
        assert all(c != "" for c in chunks)
        assert "".join(chunks).replace("\n", "") == "x" * 50

    def test_discover_advances_past_chatless_updates(self):
        """Updates without a chat id must be acked exactly one past their
        update_id, or discover re-reads them forever."""
        calls = []

        def fake_api(token, method, params=None, timeout=35):
            calls.append(dict(params or {}))
            if len(calls) == 1:
                return {"ok": True, "result": []}  # watermark probe -> 0
            if len(calls) == 2:
                return {"ok": True, "result": [{"update_id": 3}]}  # chatless
            return {"ok": True, "result": [
                {"update_id": 4, "message": {"chat": {"id": 9}}}]}

        with patch.object(telegram, "api_call", side_effect=fake_api):
            assert telegram.discover_chat_id("t", wait=30) == "9"
        assert calls[2]["offset"] == 4  # 3 + 1, exactly
