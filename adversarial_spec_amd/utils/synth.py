"""Synthetic spec generation for benchmarks and tests.

BASELINE.json configs name token counts (4k/8k/16k/32k spec) on synthetic
documents. The generator emits deterministic markdown-shaped text whose
byte-tokenizer length hits a target token count exactly.
"""

from __future__ import annotations

import random

_SECTIONS = [
    "Overview", "Goals and Non-Goals", "System Architecture", "Component Design",
    "API Design", "Data Models", "Infrastructure Requirements",
    "Security Considerations", "Error Handling Strategy", "Performance Requirements",
    "Observability", "Testing Strategy", "Deployment Strategy", "Migration Plan",
    "Open Questions",
]

_WORDS = (
    "service endpoint schema latency replica shard cache queue retry backoff "
    "token session payload index cluster failover quorum snapshot rollout canary "
    "metric alert budget throughput partition consumer producer stream batch "
    "ledger audit policy tenant region zone durability consistency checkpoint"
).split()


def synthetic_spec(target_tokens: int, seed: int = 0) -> str:
    """Markdown tech-spec-shaped text of ~target_tokens byte-tokens.

    Byte tokenizer: 1 token per byte, so the target is a byte count.
    """
    rng = random.Random(seed)
    parts: list[str] = ["# Synthetic Technical Specification\n\n"]
    size = len(parts[0])
    si = 0
    while size < target_tokens:
        header = f"## {_SECTIONS[si % len(_SECTIONS)]} ({si})\n\n"
        parts.append(header)
        size += len(header)
        para_words = [rng.choice(_WORDS) for _ in range(rng.randint(60, 120))]
        para = "The " + " ".join(para_words) + ".\n\n"
        parts.append(para)
        size += len(para)
        si += 1
    text = "".join(parts)
    return text[:target_tokens]
