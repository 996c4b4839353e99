"""In-process loopback `torch.distributed` for multi-rank tests on ONE device.

RCCL refuses two ranks on the same GPU ("Duplicate GPU detected"), so the
real multi-rank composition of the HIP path (per-rank hop geometry, striped
gather order, ring P2P, reduce-scatter of dk/dv) cannot be exercised on a
1-GPU box with real process groups.  This module fakes the small set of
torch.distributed primitives the framework uses (see grep inventory in the
test) with a thread-rendezvous store: `loopback_world(R, fn)` runs `fn(rank)`
on R python threads, each seeing `dist.get_rank()` == its own rank, and every
collective exchanges real tensors between the threads.

Backward passes must run with torch.autograd.set_multithreading_enabled(False)
(the helper does this) so collectives called from autograd execute on the
rank's own thread — otherwise the shared CUDA engine worker thread would
deadlock on the first rendezvous.
"""

from __future__ import annotations

import threading
from collections import defaultdict

import torch
import torch.distributed as dist


class FakeGroup:
    def __init__(self, ranks):
        self.ranks = tuple(ranks)

    def size(self):
        return len(self.ranks)


class FakeWork:
    def wait(self):
        return True


class _P2POp:
    def __init__(self, op, tensor, peer, group=None, tag=0):
        self.op, self.tensor, self.peer = op, tensor, peer


class LoopbackWorld:
    TIMEOUT = 120.0

    def __init__(self, world_size):
        self.world = world_size
        self.cv = threading.Condition()
        self.store = {}
        self.reads = {}
        self.tls = threading.local()
        self.failed = False

    # ---- plumbing -------------------------------------------------------
    def _ranks(self, group):
        return group.ranks if isinstance(group, FakeGroup) else tuple(range(self.world))

    def _counts(self):
        if not hasattr(self.tls, "counts"):
            self.tls.counts = defaultdict(int)
        return self.tls.counts

    def _exchange(self, ranks, value):
        """All-to-all rendezvous among `ranks`; returns {rank: value}."""
        rank = self.tls.rank
        cnt = self._counts()
        idx = cnt[("c", ranks)]
        cnt[("c", ranks)] += 1
        key = ("c", ranks, idx)
        with self.cv:
            e = self.store.setdefault(key, {})
            e[rank] = value
            if len(e) == len(ranks):
                self.cv.notify_all()
            deadline = self.TIMEOUT
            while len(self.store[key]) < len(ranks):
                if not self.cv.wait(timeout=deadline) or self.failed:
                    self.failed = True
                    raise RuntimeError(f"loopback rendezvous timeout at {key}")
            result = dict(self.store[key])
            self.reads[key] = self.reads.get(key, 0) + 1
            if self.reads[key] == len(ranks):
                del self.store[key], self.reads[key]
        return result

    def _p2p_send(self, t, dst):
        rank = self.tls.rank
        cnt = self._counts()
        idx = cnt[("s", rank, dst)]
        cnt[("s", rank, dst)] += 1
        key = ("p", rank, dst, idx)
        with self.cv:
            self.store[key] = t.detach().clone()
            self.cv.notify_all()

    def _p2p_recv(self, buf, src):
        rank = self.tls.rank
        cnt = self._counts()
        idx = cnt[("r", src, rank)]
        cnt[("r", src, rank)] += 1
        key = ("p", src, rank, idx)
        with self.cv:
            while key not in self.store:
                if not self.cv.wait(timeout=self.TIMEOUT) or self.failed:
                    self.failed = True
                    raise RuntimeError(f"loopback p2p timeout at {key}")
            val = self.store.pop(key)
        buf.copy_(val)

    # ---- faked torch.distributed API ------------------------------------
    def is_initialized(self):
        return True

    def get_rank(self, group=None):
        if isinstance(group, FakeGroup):
            return group.ranks.index(self.tls.rank)
        return self.tls.rank

    def get_world_size(self, group=None):
        if isinstance(group, FakeGroup):
            return len(group.ranks)
        return self.world

    def get_backend(self, group=None):
        return "nccl"          # exercise the RCCL fast paths

    def new_group(self, ranks=None, **kw):
        return FakeGroup(ranks if ranks is not None else range(self.world))

    def barrier(self, group=None, **kw):
        self._exchange(self._ranks(group), None)

    def all_reduce(self, tensor, op=None, group=None, async_op=False):
        vals = self._exchange(self._ranks(group), tensor.detach().clone())
        it = iter(vals.values())
        total = next(it).clone()
        for v in it:
            if op is dist.ReduceOp.MAX:
                torch.maximum(total, v, out=total)
            else:
                total += v
        tensor.copy_(total)
        return FakeWork() if async_op else None

    def all_gather(self, tensor_list, tensor, group=None, async_op=False):
        ranks = self._ranks(group)
        vals = self._exchange(ranks, tensor.detach().clone())
        for i, r in enumerate(ranks):
            tensor_list[i].copy_(vals[r])
        return FakeWork() if async_op else None

    def all_gather_into_tensor(self, out, inp, group=None, async_op=False):
        ranks = self._ranks(group)
        vals = self._exchange(ranks, inp.detach().clone())
        torch.cat([vals[r].reshape(-1) for r in ranks], out=out.view(-1))
        return FakeWork() if async_op else None

    def reduce_scatter_tensor(self, out, inp, op=None, group=None, async_op=False):
        ranks = self._ranks(group)
        vals = self._exchange(ranks, inp.detach().clone())
        it = iter(vals.values())
        total = next(it).clone()
        for v in it:
            total += v
        my = ranks.index(self.tls.rank)
        out.view(-1).copy_(total.view(len(ranks), -1)[my])
        return FakeWork() if async_op else None

    def batch_isend_irecv(self, ops):
        # deposit every send first (non-blocking), then satisfy receives
        for o in ops:
            if o.op == self.isend:
                self._p2p_send(o.tensor, o.peer)
        for o in ops:
            if o.op == self.irecv:
                self._p2p_recv(o.tensor, o.peer)
        return [FakeWork()]

    def isend(self, tensor, dst, group=None, tag=0):
        self._p2p_send(tensor, dst)
        return FakeWork()

    def irecv(self, tensor, src=None, group=None, tag=0):
        self._p2p_recv(tensor, src)
        return FakeWork()


_PATCHED = ("is_initialized", "get_rank", "get_world_size", "get_backend",
            "new_group", "barrier", "all_reduce", "all_gather",
            "all_gather_into_tensor", "reduce_scatter_tensor",
            "batch_isend_irecv", "isend", "irecv", "P2POp")


def loopback_world(world_size, fn):
    """Run `fn(rank)` on `world_size` threads under the faked dist layer.

    Returns [fn(0), fn(1), ...].  Any rank's exception fails the whole call.
    """
    lw = LoopbackWorld(world_size)
    saved = {n: getattr(dist, n) for n in _PATCHED}
    for n in _PATCHED:
        setattr(dist, n, _P2POp if n == "P2POp" else getattr(lw, n))

    results = [None] * world_size
    errors = []

    def runner(r):
        lw.tls.rank = r
        try:
            with torch.autograd.set_multithreading_enabled(False):
                results[r] = fn(r)
        except BaseException as e:          # noqa: BLE001 — surface to caller
            errors.append((r, e))
            lw.failed = True
            with lw.cv:
                lw.cv.notify_all()

    threads = [threading.Thread(target=runner, args=(r,), daemon=True)
               for r in range(world_size)]
    try:
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=LoopbackWorld.TIMEOUT * 4)
            if t.is_alive():
                lw.failed = True
                errors.append((-1, RuntimeError("loopback thread join timeout")))
                break
    finally:
        for n, v in saved.items():
            setattr(dist, n, v)
        from ring_attention_amd.parallel import topology
        topology._RING_GROUPS.clear()
    if errors:
        raise errors[0][1]
    return results
