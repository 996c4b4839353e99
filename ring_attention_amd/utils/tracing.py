"""Lightweight tracing/observability hooks (the reference had none — SURVEY.md §5).

- ``trace_range(name)``: context manager emitting a roctx range (visible in
  rocprofv3 --marker-trace timelines) plus a torch.profiler record_function
  scope, when those facilities exist; free no-ops otherwise.
- ``RingStats``: per-process counters for ring communication (hops, bytes,
  comm wall time) updated by the ring engine itself (``all_ring_pass``,
  ``ring_pass`` multi-hop routing and ``RingAccumulator``), so every ring
  attention strategy feeds them; read/reset via GLOBAL_RING_STATS.
"""

from __future__ import annotations

import contextlib
import time
from dataclasses import dataclass, field

import torch

try:  # roctx via torch's bundled roctracer bindings (ROCm builds)
    from torch._C import _roctx  # type: ignore[attr-defined]
    _HAVE_ROCTX = True
except Exception:
    _roctx = None
    _HAVE_ROCTX = False


@contextlib.contextmanager
def trace_range(name: str):
    """Named range for profilers: roctx (rocprofv3) + torch.profiler."""
    if _HAVE_ROCTX:
        _roctx.rangePushA(name)
    with torch.profiler.record_function(name):
        try:
            yield
        finally:
            if _HAVE_ROCTX:
                _roctx.rangePop()


@dataclass
class RingStats:
    """Cumulative ring-communication counters (per process)."""
    hops: int = 0
    bytes_sent: int = 0
    wall_s: float = 0.0
    _t0: float = field(default=0.0, repr=False)

    def start(self):
        self._t0 = time.perf_counter()

    def stop(self, hops: int, bytes_sent: int):
        self.wall_s += time.perf_counter() - self._t0
        self.hops += hops
        self.bytes_sent += bytes_sent

    def reset(self):
        self.hops = 0
        self.bytes_sent = 0
        self.wall_s = 0.0


GLOBAL_RING_STATS = RingStats()
