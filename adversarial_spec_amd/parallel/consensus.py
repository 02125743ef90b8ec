"""Opponent-parallel consensus over RCCL/xGMI.

The reference gathers N opponent responses with `as_completed` over threads
(models.py:699-722) and reduces consensus with a Python `all()`
(debate.py:853). Distributed MI355X equivalent:

  - one process per GPU (torch.distributed, backend "nccl" == RCCL on
    ROCm; "gloo" for CPU tests), one opponent per rank;
  - the round barrier is ONE fused all-gather of a fixed-size int32
    buffer per rank: [agreed, error, n_tokens, reserved | token ids...].
    Messages are tiny (<=8000 tokens ~ 32 KB) so the gather is
    latency-bound: one fused collective, never per-token traffic — xGMI
    is 7 p2p links x ~153 GB/s and a ring all-gather is per-link bound;
  - the all-agreed reduction rides in the header word (C2), no second
    collective;
  - `gather_async` launches the collective on a dedicated stream so the
    next round's prefill can overlap it (the BASELINE north star).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist

HDR = 4  # [agreed, error, n_tokens, reserved]


@dataclass
class RankResult:
    rank: int
    agreed: bool
    error: bool
    token_ids: list[int]


def pack_result(
    token_ids: list[int],
    agreed: bool,
    error: bool,
    max_tokens: int,
    device: torch.device,
) -> torch.Tensor:
    """Pack one opponent's round result into the fixed gather buffer."""
    buf = torch.zeros(HDR + max_tokens, dtype=torch.int32, device=device)
    n = min(len(token_ids), max_tokens)
    buf[0] = int(agreed)
    buf[1] = int(error)
    buf[2] = n
    if n:
        buf[HDR : HDR + n] = torch.tensor(token_ids[:n], dtype=torch.int32, device=device)
    return buf


def unpack_results(gathered: torch.Tensor) -> list[RankResult]:
    """Inverse of pack_result over the [world, HDR+max] gathered tensor."""
    out = []
    g = gathered.cpu()
    for rank in range(g.shape[0]):
        row = g[rank]
        n = int(row[2].item())
        out.append(
            RankResult(
                rank=rank,
                agreed=bool(row[0].item()),
                error=bool(row[1].item()),
                token_ids=row[HDR : HDR + n].tolist(),
            )
        )
    return out


def gather_round(
    token_ids: list[int],
    agreed: bool,
    error: bool,
    max_tokens: int = 8192,
    group: Optional[dist.ProcessGroup] = None,
    device: Optional[torch.device] = None,
) -> tuple[list[RankResult], bool]:
    """Synchronous fused all-gather of every rank's critique.

    Returns (per-rank results, all_agreed) where all_agreed follows the
    reference semantics: errored opponents are excluded; False when every
    opponent errored (reference: debate.py:845-853).
    """
    if device is None:
        backend = dist.get_backend(group)
        device = torch.device("cuda") if backend == "nccl" else torch.device("cpu")
    world = dist.get_world_size(group)
    mine = pack_result(token_ids, agreed, error, max_tokens, device)
    out = torch.zeros(world * (HDR + max_tokens), dtype=torch.int32, device=device)
    dist.all_gather_into_tensor(out, mine, group=group)
    results = unpack_results(out.view(world, HDR + max_tokens))
    ok = [r for r in results if not r.error]
    all_agreed = bool(ok) and all(r.agreed for r in ok)
    return results, all_agreed


class AsyncRoundGather:
    """All-gather on a dedicated stream, overlapping the next prefill.

    Usage on the nccl(RCCL) backend:
        g = AsyncRoundGather(max_tokens)
        g.launch(token_ids, agreed, error)    # returns immediately
        ... start round k+1 prefill on the default stream ...
        results, all_agreed = g.wait()

    On gloo/CPU the collective runs inline (no streams) — same API.
    """

    def __init__(self, max_tokens: int = 8192,
                 group: Optional[dist.ProcessGroup] = None) -> None:
        self.max_tokens = max_tokens
        self.group = group
        backend = dist.get_backend(group)
        self.is_nccl = backend == "nccl"
        self.device = torch.device("cuda") if self.is_nccl else torch.device("cpu")
        self.stream = torch.cuda.Stream() if self.is_nccl else None
        self._out: Optional[torch.Tensor] = None
        self._work = None

    def launch(self, token_ids: list[int], agreed: bool, error: bool) -> None:
        world = dist.get_world_size(self.group)
        mine = pack_result(token_ids, agreed, error, self.max_tokens, self.device)
        self._world = world
        self._out = torch.zeros(
            world * (HDR + self.max_tokens), dtype=torch.int32, device=self.device
        )
        if self.is_nccl:
            # make the comm stream wait for the decode that produced `mine`
            self.stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self.stream):
                self._work = dist.all_gather_into_tensor(
                    self._out, mine, group=self.group, async_op=True
                )
        else:
            self._work = dist.all_gather_into_tensor(
                self._out, mine, group=self.group, async_op=True
            )

    def wait(self) -> tuple[list[RankResult], bool]:
        assert self._work is not None, "launch() before wait()"
        self._work.wait()
        if self.is_nccl:
            torch.cuda.current_stream().wait_stream(self.stream)
        results = unpack_results(self._out.view(self._world, HDR + self.max_tokens))
        ok = [r for r in results if not r.error]
        all_agreed = bool(ok) and all(r.agreed for r in ok)
        self._work = None
        return results, all_agreed
