"""Per-phase wall-clock timing (prefill / decode / gather).

The reference has no tracing at all; BASELINE requires
per-phase timings and rocprof-visible kernel names. GPU phases synchronize
the device so a phase's time is real, not launch time.
"""

from __future__ import annotations

import time
from contextlib import contextmanager
from typing import Iterator


class PhaseTimer:
    def __init__(self, sync_cuda: bool = False) -> None:
        self.sync_cuda = sync_cuda
        self.ms: dict[str, float] = {}
        self.counts: dict[str, int] = {}

    def _sync(self) -> None:
        if self.sync_cuda:
            import torch

            if torch.cuda.is_available():
                # Sync only this thread's stream: co-resident opponents run
                # concurrently on per-engine streams, and a device-wide
                # synchronize here would serialize them at phase boundaries.
                torch.cuda.current_stream().synchronize()

    @contextmanager
    def phase(self, name: str) -> Iterator[None]:
        self._sync()
        t0 = time.perf_counter()
        try:
            yield
        finally:
            self._sync()
            dt = (time.perf_counter() - t0) * 1000.0
            self.ms[name] = self.ms.get(name, 0.0) + dt
            self.counts[name] = self.counts.get(name, 0) + 1

    def as_dict(self) -> dict[str, float]:
        return dict(self.ms)
