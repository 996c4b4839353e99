"""The ring transport: double-buffered neighbor exchange over torch.distributed P2P.

Capability parity with the reference's ring pass
(/root/reference/ring_attention_pytorch/ring.py:51-124) but re-designed for
RCCL over xGMI:

- Each hop is ONE grouped batch_isend_irecv (send right / receive left).  On
  an MI355X node this is a single-xGMI-link transfer (~153 GB/s per hop).
- The exchange for hop t+1 is POSTED before hop t's compute is consumed, so
  under RCCL the transfer proceeds on RCCL's internal streams while the
  attention kernel for the current hop runs: communication hides behind
  compute.  The reference instead did send -> wait -> barrier -> compute,
  fully serialized (ring.py:51-60).
- No per-hop ``dist.barrier()``: buffer alternation plus waiting a hop's own
  requests before reusing its buffers is sufficient for correctness.
- Multi-hop routing (``ring_pass(..., num_hops=k)``) is implemented correctly;
  the reference declared but ignored ``num_ring_passes`` (ring.py:62-77),
  breaking dk/dv homecoming for lookback-truncated rings.
"""

from __future__ import annotations

from typing import Iterator, NamedTuple, Sequence

import torch
import torch.distributed as dist
from torch import Tensor

from .topology import RingTopology, is_distributed


class RingInfo(NamedTuple):
    hop: int                 # 0-based pass index
    source_ring_rank: int    # ring rank whose original shard we hold this hop
    is_first: bool
    is_last: bool


def _exchange(topo: RingTopology, tensors: Sequence[Tensor], recv_bufs: Sequence[Tensor]):
    """Post one grouped send-right/recv-left for every tensor; return reqs."""
    ops = []
    for send, recv in zip(tensors, recv_bufs):
        ops.append(dist.P2POp(dist.isend, send, topo.right))
        ops.append(dist.P2POp(dist.irecv, recv, topo.left))
    return dist.batch_isend_irecv(ops)


def one_ring_pass(topo: RingTopology, *tensors: Tensor) -> tuple[Tensor, ...]:
    """Synchronously hop every tensor one step rightward around the ring."""
    out = ring_pass(topo, *tensors, num_hops=1)
    return out


def ring_pass(topo: RingTopology, *tensors: Tensor, num_hops: int = 1) -> tuple[Tensor, ...]:
    """Hop every tensor ``num_hops`` steps rightward; returns the received tensors.

    Unlike the reference, multiple hops genuinely perform multiple exchanges
    (used for dk/dv homecoming when lookback truncates the ring walk).
    """
    if topo.ring_size == 1 or not is_distributed():
        return tuple(tensors)
    from ..utils.tracing import GLOBAL_RING_STATS, trace_range
    # NEVER recv into the caller's tensors: ping-pong between two INTERNAL
    # buffers (reusing the inputs as receive targets silently mutated them
    # from hop 2 on — the world>=3 saved-KV corruption found in round 2)
    cur = [t.contiguous() for t in tensors]
    recv = [torch.empty_like(t) for t in cur]
    spare = None
    nbytes = sum(t.element_size() * t.numel() for t in cur)
    GLOBAL_RING_STATS.start()
    hops_done = 0
    with trace_range("ring_pass.multi_hop"):
        for hop in range(num_hops % topo.ring_size):  # ring_size hops = identity
            reqs = _exchange(topo, cur, recv)
            for r in reqs:
                r.wait()
            if hop == 0:
                spare = [torch.empty_like(t) for t in cur]  # caller's cur retired
            else:
                spare = cur
            cur, recv = recv, spare
            hops_done += 1
    GLOBAL_RING_STATS.stop(hops_done, hops_done * nbytes)
    return tuple(cur)


def all_ring_pass(
    topo: RingTopology,
    *tensors: Tensor,
    max_hops: int | None = None,
) -> Iterator[tuple[RingInfo, tuple[Tensor, ...]]]:
    """Drive ``max_hops`` (default ring_size) passes of ``tensors`` around the ring.

    Yields ``(RingInfo, tensors)`` per hop.  The next hop's exchange is posted
    BEFORE yielding the current hop's tensors, so the transfer overlaps
    whatever compute the consumer enqueues.  The consumer must not mutate the
    yielded tensors (they are in flight).
    """
    max_hops = topo.ring_size if max_hops is None else min(max_hops, topo.ring_size)
    assert max_hops >= 1

    if topo.ring_size == 1 or not is_distributed() or max_hops == 1:
        yield RingInfo(0, topo.ring_rank, True, max_hops == 1), tensors
        # even with a single compute hop on a real ring, nothing needs sending
        return

    from ..utils.tracing import GLOBAL_RING_STATS, trace_range
    # Buffer discipline: the caller's tensors are SEND-only.  Receives go to
    # internal buffers that ping-pong between two sets allocated here; the
    # input tensors are never written (reusing them as receive targets from
    # hop 2 on silently mutated the caller's K/V — the world>=3 corruption
    # of saved forward tensors found in round 2; world 2 never re-used them,
    # which is why every 2-rank test passed).
    cur = [t.contiguous() for t in tensors]
    recv = [torch.empty_like(t) for t in cur]
    spare = None
    reqs = None
    nbytes = sum(t.element_size() * t.numel() for t in cur)

    for hop in range(max_hops):
        is_last = hop == max_hops - 1
        if not is_last:
            # post the exchange for the NEXT hop now; compute on `cur` overlaps it
            with trace_range(f"ring_pass.post_hop{hop + 1}"):
                reqs = _exchange(topo, cur, recv)
            GLOBAL_RING_STATS.start()

        yield RingInfo(hop, topo.source_of_hop(hop), hop == 0, is_last), tuple(cur)

        if not is_last:
            with trace_range(f"ring_pass.wait_hop{hop + 1}"):
                for r in reqs:
                    r.wait()
            GLOBAL_RING_STATS.stop(1, nbytes)
            if hop == 0:
                spare = [torch.empty_like(t) for t in cur]  # caller's cur retired
            else:
                spare = cur
            cur, recv = recv, spare


def null_ring_pass(*tensors: Tensor) -> Iterator[tuple[RingInfo, tuple[Tensor, ...]]]:
    """Degenerate single-yield iterator for non-distributed execution."""
    yield RingInfo(0, 0, True, True), tensors


class RingAccumulator:
    """Pipelined ring accumulation for backward's circulating dk/dv.

    Shard s's gradient accumulates contributions from every rank it visits.
    The naive scheme (receive acc -> compute -> add -> send) puts the transfer
    on the critical path; here each rank computes its LOCAL contribution first
    (which only needs the prefetched k/v, not the incoming accumulator), then
    waits for the incoming accumulator, adds, and posts the send — so the
    accumulator transfer for hop t overlaps the attention-backward compute of
    hop t+1.

    Usage, per hop t = 0..P-1::

        contribution = <backward kernel for the shard held this hop>
        acc.step(contribution, is_last = t == P-1)
    then::
        dkv_home = acc.finish(total_hops=P)
    """

    def __init__(self, topo: RingTopology):
        self.topo = topo
        self._acc: Tensor | None = None
        self._recv: Tensor | None = None
        self._reqs = None
        self._distributed = topo.ring_size > 1 and is_distributed()

    def step(self, contribution: Tensor, is_last: bool):
        if not self._distributed:
            if self._acc is None:
                self._acc = contribution
            else:  # single-rank multi-hop cannot happen; defensive
                self._acc = self._acc + contribution
            return

        if self._reqs is not None:
            from ..utils.tracing import GLOBAL_RING_STATS
            for r in self._reqs:
                r.wait()
            self._reqs = None
            GLOBAL_RING_STATS.stop(
                1, self._recv.element_size() * self._recv.numel())
            contribution = contribution + self._recv

        self._acc = contribution.contiguous()
        if not is_last:
            from ..utils.tracing import GLOBAL_RING_STATS
            self._recv = torch.empty_like(self._acc)
            self._reqs = _exchange(self.topo, [self._acc], [self._recv])
            GLOBAL_RING_STATS.start()

    def finish(self, total_hops: int) -> Tensor:
        """Route the final accumulator to its home rank and return it.

        After ``total_hops`` compute hops, rank r holds the accumulator for
        the shard of ring rank ``r - (total_hops - 1)``; it needs
        ``ring_size - (total_hops - 1)`` more rightward hops to reach home
        (mod ring_size — one hop in the untruncated case).
        """
        assert self._acc is not None
        if not self._distributed:
            return self._acc
        remaining = (self.topo.ring_size - (total_hops - 1)) % self.topo.ring_size
        (home,) = ring_pass(self.topo, self._acc, num_hops=remaining)
        return home
