"""Tokenizers for the on-node engine.

Two implementations behind one interface:

- ByteTokenizer: deterministic byte-level tokenizer (ids 0-255 = raw bytes)
  with Llama-3-style special-token ids parked at 128000+. Needs no
  downloaded vocab files, round-trips arbitrary UTF-8, and gives EXACT token
  counts (replacing the reference's len//4 estimate, models.py:444-447).
  This is the default for random-init opponents and synthetic benchmarks.

- HFTokenizer: wraps a transformers tokenizer loaded from a local path when
  a registry alias pins real weights.

The chat template is Llama-3 shaped (header/eot specials around
system/user/assistant turns).
"""

from __future__ import annotations

from typing import Optional, Protocol

# Llama-3 special-token layout (ids relative to a 128256 vocab).
BOS_ID = 128000
EOS_ID = 128001
START_HEADER_ID = 128006
END_HEADER_ID = 128007
EOT_ID = 128009

_SPECIAL_NAMES = {
    BOS_ID: "<|begin_of_text|>",
    EOS_ID: "<|end_of_text|>",
    START_HEADER_ID: "<|start_header_id|>",
    END_HEADER_ID: "<|end_header_id|>",
    EOT_ID: "<|eot_id|>",
}


class Tokenizer(Protocol):
    vocab_size: int
    bos_id: int
    eos_id: int
    eot_id: int

    def encode(self, text: str) -> list[int]: ...
    def decode(self, ids: list[int]) -> str: ...
    def render_chat(self, system: str, user: str) -> list[int]: ...


class ByteTokenizer:
    """Byte-level tokenizer: id i in [0,256) is byte i; specials at 128000+.

    For vocabularies smaller than 128256 (test configs), specials are
    remapped to the top of the vocab while keeping the same ordering.
    """

    def __init__(self, vocab_size: int = 128256) -> None:
        if vocab_size < 256 + 16:
            raise ValueError("vocab_size must be >= 272 for byte ids + specials")
        self.vocab_size = vocab_size
        if vocab_size > 128009:
            base = 128000
        else:
            base = vocab_size - 16
        self.bos_id = base + (BOS_ID - 128000)
        self.eos_id = base + (EOS_ID - 128000)
        self.start_header_id = base + (START_HEADER_ID - 128000)
        self.end_header_id = base + (END_HEADER_ID - 128000)
        self.eot_id = base + (EOT_ID - 128000)
        self._special_names = {
            self.bos_id: "<|begin_of_text|>",
            self.eos_id: "<|end_of_text|>",
            self.start_header_id: "<|start_header_id|>",
            self.end_header_id: "<|end_header_id|>",
            self.eot_id: "<|eot_id|>",
        }

    def encode(self, text: str) -> list[int]:
        return list(text.encode("utf-8"))

    def decode(self, ids: list[int]) -> str:
        out: list[str] = []
        buf = bytearray()
        for i in ids:
            if 0 <= i < 256:
                buf.append(i)
            else:
                if buf:
                    out.append(buf.decode("utf-8", errors="replace"))
                    buf = bytearray()
                name = self._special_names.get(i)
                if name:
                    out.append(name)
                # other out-of-range ids (random-init models emit them) are
                # dropped silently — they carry no text.
        if buf:
            out.append(buf.decode("utf-8", errors="replace"))
        return "".join(out)

    def _header(self, role: str) -> list[int]:
        return (
            [self.start_header_id]
            + self.encode(role)
            + [self.end_header_id]
            + self.encode("\n\n")
        )

    def render_chat(self, system: str, user: str) -> list[int]:
        ids = [self.bos_id]
        ids += self._header("system") + self.encode(system) + [self.eot_id]
        ids += self._header("user") + self.encode(user) + [self.eot_id]
        ids += self._header("assistant")
        return ids

    def stop_ids(self) -> set[int]:
        return {self.eos_id, self.eot_id}


class HFTokenizer:
    """transformers tokenizer loaded from a local directory (no network)."""

    def __init__(self, path: str) -> None:
        from transformers import AutoTokenizer

        self._tok = AutoTokenizer.from_pretrained(path, local_files_only=True)
        self.vocab_size = len(self._tok)
        self.bos_id = self._tok.bos_token_id or BOS_ID
        self.eos_id = self._tok.eos_token_id or EOS_ID
        self.eot_id = self._tok.convert_tokens_to_ids("<|eot_id|>")
        if self.eot_id is None or self.eot_id < 0:
            self.eot_id = self.eos_id

    def encode(self, text: str) -> list[int]:
        return self._tok.encode(text, add_special_tokens=False)

    def decode(self, ids: list[int]) -> str:
        return self._tok.decode(ids, skip_special_tokens=False)

    def render_chat(self, system: str, user: str) -> list[int]:
        try:
            return self._tok.apply_chat_template(
                [
                    {"role": "system", "content": system},
                    {"role": "user", "content": user},
                ],
                add_generation_prompt=True,
            )
        except Exception:
            text = f"{system}\n\n{user}\n\n"
            return ([self.bos_id] if self.bos_id is not None else []) + self.encode(text)

    def stop_ids(self) -> set[int]:
        return {self.eos_id, self.eot_id}


def build_tokenizer(vocab_size: int, weights_path: Optional[str] = None) -> Tokenizer:
    """HF tokenizer when real weights are pinned, byte tokenizer otherwise."""
    if weights_path:
        try:
            return HFTokenizer(weights_path)
        except Exception:
            pass
    return ByteTokenizer(vocab_size)
