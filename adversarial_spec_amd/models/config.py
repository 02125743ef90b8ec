"""Model architecture configs for the on-node engine.

Llama-family decoder-only transformers (RMSNorm + RoPE + GQA + SwiGLU).
Presets cover the BASELINE.json opponents: Llama-3-8B (configs 2-4),
Llama-3-70B (config 5, TP=8), Mistral-7B (heterogeneous config 4), plus a
4-layer `tiny` config for CPU tests and smoke.
"""

from __future__ import annotations

from dataclasses import dataclass


@dataclass(frozen=True)
class LlamaConfig:
    name: str
    dim: int
    n_layers: int
    n_heads: int
    n_kv_heads: int
    ffn_dim: int
    vocab_size: int
    max_seq_len: int = 8192
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5
    # explicit head_dim survives tensor-parallel sharding (where n_heads
    # shrinks but the per-head width does not); None derives dim/n_heads.
    head_dim_override: int = 0

    @property
    def head_dim(self) -> int:
        return self.head_dim_override or self.dim // self.n_heads

    def param_count(self) -> int:
        """Approximate parameter count (embeddings untied)."""
        per_layer = (
            self.dim * (self.n_heads + 2 * self.n_kv_heads) * self.head_dim  # qkv
            + self.dim * self.dim  # o
            + 3 * self.dim * self.ffn_dim  # gate, up, down
            + 2 * self.dim  # norms
        )
        return (
            2 * self.vocab_size * self.dim  # embed + lm_head
            + self.n_layers * per_layer
            + self.dim
        )


PRESETS: dict[str, LlamaConfig] = {
    "llama-3-8b": LlamaConfig(
        name="llama-3-8b", dim=4096, n_layers=32, n_heads=32, n_kv_heads=8,
        ffn_dim=14336, vocab_size=128256, max_seq_len=32768, rope_theta=500000.0,
    ),
    "llama-3-70b": LlamaConfig(
        name="llama-3-70b", dim=8192, n_layers=80, n_heads=64, n_kv_heads=8,
        ffn_dim=28672, vocab_size=128256, max_seq_len=32768, rope_theta=500000.0,
    ),
    "mistral-7b": LlamaConfig(
        name="mistral-7b", dim=4096, n_layers=32, n_heads=32, n_kv_heads=8,
        ffn_dim=14336, vocab_size=32000, max_seq_len=32768, rope_theta=10000.0,
    ),
    # CPU-testable / smoke model: byte tokenizer fits in 1024 vocab.
    "tiny": LlamaConfig(
        name="tiny", dim=256, n_layers=4, n_heads=8, n_kv_heads=2,
        ffn_dim=688, vocab_size=1024, max_seq_len=2048, rope_theta=10000.0,
    ),
    # A mid-size config for single-GPU kernel shakeout (hd=128 MFMA path).
    "debug-1b": LlamaConfig(
        name="debug-1b", dim=2048, n_layers=16, n_heads=16, n_kv_heads=8,
        ffn_dim=8192, vocab_size=128256, max_seq_len=32768, rope_theta=500000.0,
    ),
}


def get_config(arch: str) -> LlamaConfig:
    if arch not in PRESETS:
        raise ValueError(f"Unknown architecture '{arch}'. Known: {', '.join(PRESETS)}")
    return PRESETS[arch]
