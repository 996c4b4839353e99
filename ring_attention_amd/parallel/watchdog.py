"""Failure detection for distributed runs (the reference had none: a hung
peer deadlocked `req.wait()` forever — SURVEY.md §5).

Two mechanisms:

- ``init_distributed(...)``: process-group init with an explicit collective
  TIMEOUT so a dead peer surfaces as an error instead of an infinite hang
  (NCCL/RCCL honors the timeout via its async error handling; gloo enforces
  it directly).
- ``Watchdog``: a monitor thread that fires a callback (default: log + dump
  stack traces) when the training loop stops making progress — catches hangs
  above the collective layer too.
"""

from __future__ import annotations

import datetime
import faulthandler
import os
import sys
import threading
import time

import torch.distributed as dist


def init_distributed(backend: str | None = None, timeout_s: float = 600.0, **kwargs):
    """init_process_group with a real timeout and RCCL async error handling."""
    import torch
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    # RCCL: surface collective timeouts as exceptions instead of hanging
    os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "1")
    dist.init_process_group(backend, timeout=datetime.timedelta(seconds=timeout_s),
                            **kwargs)


class Watchdog:
    """Fires ``on_stall`` if ``tick()`` is not called within ``stall_s``.

    Usage::

        wd = Watchdog(stall_s=120)
        wd.start()
        for step in loop:
            ...
            wd.tick(step)
        wd.stop()
    """

    def __init__(self, stall_s: float = 120.0, on_stall=None, check_every_s: float = 5.0):
        self.stall_s = stall_s
        self.check_every_s = check_every_s
        self.on_stall = on_stall or self._default_on_stall
        self._last = time.monotonic()
        self._last_step = -1
        self._stop = threading.Event()
        self._thread: threading.Thread | None = None
        self.stalled = False

    def _default_on_stall(self, last_step: int, elapsed: float):
        rank = dist.get_rank() if dist.is_initialized() else 0
        sys.stderr.write(
            f"[ring_attention_amd watchdog] rank {rank}: no progress for "
            f"{elapsed:.0f}s (last step {last_step}); dumping stacks\n")
        faulthandler.dump_traceback(file=sys.stderr)

    def tick(self, step: int | None = None):
        self._last = time.monotonic()
        if step is not None:
            self._last_step = step
        self.stalled = False

    def _run(self):
        while not self._stop.wait(self.check_every_s):
            elapsed = time.monotonic() - self._last
            if elapsed > self.stall_s and not self.stalled:
                self.stalled = True
                self.on_stall(self._last_step, elapsed)

    def start(self):
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="ring-attn-watchdog")
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=self.check_every_s + 1)

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()
