"""Telegram human-in-the-loop channel (reference: telegram_bot.py).

Raw Bot-API client over urllib (no SDK): chunked sends at the API's 4096
character limit, long-polling for a human reply, chat-id discovery, and a
standalone CLI (`python -m adversarial_spec_amd.telegram setup|send|poll|notify`).
Host-side only — no GPU involvement.
"""

from __future__ import annotations

import json
import os
import sys
import time
import urllib.error
import urllib.parse
import urllib.request
from typing import Any, Optional

TELEGRAM_API = "https://api.telegram.org/bot{token}/{method}"
MAX_MESSAGE_LENGTH = 4096
CHUNK_SLEEP = 0.5  # seconds between chunks (reference: telegram_bot.py:155)
LONG_POLL_SLICE = 30  # max seconds per getUpdates long-poll (reference: :197)


def get_config() -> tuple[Optional[str], Optional[str]]:
    """(bot token, chat id) from TELEGRAM_BOT_TOKEN / TELEGRAM_CHAT_ID."""
    return os.environ.get("TELEGRAM_BOT_TOKEN"), os.environ.get("TELEGRAM_CHAT_ID")


def api_call(token: str, method: str, params: Optional[dict] = None,
             timeout: int = 35) -> dict[str, Any]:
    """POST a Bot-API method; raise RuntimeError on transport/API failure."""
    url = TELEGRAM_API.format(token=token, method=method)
    data = urllib.parse.urlencode(params or {}).encode()
    req = urllib.request.Request(url, data=data)
    try:
        with urllib.request.urlopen(req, timeout=timeout) as resp:
            payload = json.loads(resp.read().decode())
    except urllib.error.HTTPError as e:
        raise RuntimeError(f"Telegram API HTTP error: {e.code}") from e
    except urllib.error.URLError as e:
        raise RuntimeError(f"Telegram API unreachable: {e.reason}") from e
    if not payload.get("ok"):
        raise RuntimeError(f"Telegram API error: {payload.get('description')}")
    return payload


def send_message(token: str, chat_id: str, text: str) -> bool:
    try:
        api_call(token, "sendMessage", {"chat_id": chat_id, "text": text})
        return True
    except RuntimeError as e:
        print(f"Telegram send failed: {e}", file=sys.stderr)
        return False


def split_message(text: str, limit: int = MAX_MESSAGE_LENGTH) -> list[str]:
    """Split text into <=limit chunks, preferring newline boundaries."""
    if len(text) <= limit:
        return [text]
    chunks: list[str] = []
    rest = text
    while len(rest) > limit:
        cut = rest.rfind("\n", 0, limit)
        if cut <= 0:
            cut = limit
        chunks.append(rest[:cut])
        rest = rest[cut:].lstrip("\n")
    if rest:
        chunks.append(rest)
    return chunks


def send_long_message(token: str, chat_id: str, text: str) -> bool:
    """Chunked send with an inter-chunk pause to respect rate limits."""
    chunks = split_message(text)
    for i, chunk in enumerate(chunks):
        if not send_message(token, chat_id, chunk):
            return False
        if i < len(chunks) - 1:
            time.sleep(CHUNK_SLEEP)
    return True


def get_last_update_id(token: str) -> int:
    """Highest update_id currently queued (0 when none)."""
    try:
        payload = api_call(token, "getUpdates", {"timeout": 0})
    except RuntimeError:
        return 0
    updates = payload.get("result", [])
    return max((u.get("update_id", 0) for u in updates), default=0)


def poll_for_reply(token: str, chat_id: str, after_update_id: int,
                   timeout: int = 60) -> Optional[str]:
    """Long-poll for the next text message in chat_id after a watermark.

    Slices the overall timeout into <=30s getUpdates long-polls; filters on
    chat id; acknowledges consumed updates via the offset parameter.
    """
    deadline = time.time() + timeout
    offset = after_update_id + 1
    while time.time() < deadline:
        remaining = max(1, int(deadline - time.time()))
        poll = min(LONG_POLL_SLICE, remaining)
        try:
            payload = api_call(
                token, "getUpdates", {"timeout": poll, "offset": offset},
                timeout=poll + 5,
            )
        except RuntimeError:
            return None
        for update in payload.get("result", []):
            offset = max(offset, update.get("update_id", 0) + 1)
            msg = update.get("message") or {}
            if str((msg.get("chat") or {}).get("id")) == str(chat_id) and msg.get("text"):
                # ack consumed updates
                try:
                    api_call(token, "getUpdates", {"timeout": 0, "offset": offset})
                except RuntimeError:
                    pass
                return msg["text"]
    return None


def discover_chat_id(token: str, wait: int = 60) -> Optional[str]:
    """Wait for any incoming message and report its chat id (setup flow)."""
    print("Send any message to your bot now; waiting for it...", file=sys.stderr)
    deadline = time.time() + wait
    offset = get_last_update_id(token) + 1
    while time.time() < deadline:
        try:
            payload = api_call(token, "getUpdates",
                               {"timeout": min(LONG_POLL_SLICE, int(deadline - time.time()) or 1),
                                "offset": offset})
        except RuntimeError:
            return None
        for update in payload.get("result", []):
            offset = max(offset, update.get("update_id", 0) + 1)
            chat = (update.get("message") or {}).get("chat") or {}
            if chat.get("id") is not None:
                return str(chat["id"])
    return None


# ---------------------------------------------------------------------------
# Standalone CLI (reference: telegram_bot.py:404-439)
# ---------------------------------------------------------------------------

def main(argv: Optional[list[str]] = None) -> int:
    import argparse

    parser = argparse.ArgumentParser(description="Telegram channel utility")
    parser.add_argument("command", choices=["setup", "send", "poll", "notify"])
    parser.add_argument("--message", help="Message text (send/notify)")
    parser.add_argument("--timeout", type=int, default=60, help="Poll timeout seconds")
    args = parser.parse_args(argv)

    token, chat_id = get_config()
    if not token:
        print("Error: TELEGRAM_BOT_TOKEN not set", file=sys.stderr)
        return 2

    if args.command == "setup":
        found = discover_chat_id(token, wait=args.timeout)
        if found:
            print(f"Chat ID: {found}")
            print(f"Export it:  export TELEGRAM_CHAT_ID={found}")
            return 0
        print("No message received.", file=sys.stderr)
        return 1

    if not chat_id:
        print("Error: TELEGRAM_CHAT_ID not set (run setup first)", file=sys.stderr)
        return 2

    if args.command in ("send", "notify"):
        text = args.message or sys.stdin.read()
        ok = send_long_message(token, chat_id, text)
        return 0 if ok else 1

    if args.command == "poll":
        watermark = get_last_update_id(token)
        reply = poll_for_reply(token, chat_id, watermark, timeout=args.timeout)
        if reply:
            print(reply)
            return 0
        return 1
    return 1  # unreachable: argparse choices covers every command


if __name__ == "__main__":
    sys.exit(main())
