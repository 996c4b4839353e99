"""Ring topology math: rank/world helpers and sub-ring neighbor arithmetic.

Capability parity with the reference's distributed helpers and neighbor math
(/root/reference/ring_attention_pytorch/distributed.py:31-41,
 /root/reference/ring_attention_pytorch/ring.py:35-47), re-designed for an
MI355X node: world is one process per GPU over RCCL ("nccl" backend on ROCm);
``ring_size < world_size`` splits the world into independent sub-rings so one
node can run several data-parallel rings over xGMI simultaneously.
"""

from __future__ import annotations

import functools

import torch.distributed as dist

# cache: (default-group id, world_size, ring_size) -> list of per-ring process
# groups.  Keyed on the default group's identity so a destroy_process_group +
# re-init with the same sizes does not return stale dead group handles.
_RING_GROUPS: dict[tuple[int, int, int], list] = {}


def _default_group_id() -> int:
    if not dist.is_initialized():
        return 0
    try:
        return id(dist.distributed_c10d._get_default_group())
    except Exception:
        # fake/monkey-patched dist (tests) may report initialized without a
        # real default group — treat as one stable epoch
        return 0


def init_ring_groups(ring_size: int) -> None:
    """Eagerly create the sub-ring process groups for ``ring_size``.

    ``dist.new_group`` is itself a collective: EVERY rank must create EVERY
    ring's group in the same order relative to all other collectives.  Call
    this once right after ``init_process_group`` (before any forward) when
    using ``ring_size < world_size`` — ``RingTopology`` also triggers it at
    construction (the start of the first forward), but an explicit startup
    call is the robust pattern for complex training loops that may issue
    other collectives concurrently.
    """
    world = get_world_size()
    if ring_size >= world or not dist.is_initialized():
        return
    key = (_default_group_id(), world, ring_size)
    if key not in _RING_GROUPS:
        groups = []
        for ring in range(world // ring_size):
            ranks = list(range(ring * ring_size, (ring + 1) * ring_size))
            groups.append(dist.new_group(ranks))
        _RING_GROUPS[key] = groups


def is_distributed() -> bool:
    return dist.is_initialized() and dist.get_world_size() > 1


@functools.lru_cache(maxsize=None)
def _cached_rank_world() -> tuple[int, int]:
    if not dist.is_initialized():
        return 0, 1
    return dist.get_rank(), dist.get_world_size()


def get_rank() -> int:
    # NOTE: not lru_cached on its own so tests that tear down/re-init process
    # groups of different sizes in one process stay correct.
    if not dist.is_initialized():
        return 0
    return dist.get_rank()


def get_world_size() -> int:
    if not dist.is_initialized():
        return 1
    return dist.get_world_size()


class RingTopology:
    """Neighbor math for a (sub-)ring.

    The world of size W is partitioned into ``W // ring_size`` independent
    rings of ``ring_size`` consecutive ranks; ring ``i`` owns global ranks
    ``[i*ring_size, (i+1)*ring_size)``.  ``ring_rank`` is the position within
    the ring.  With ``ring_size == world_size`` this is the plain full ring.
    """

    def __init__(self, ring_size: int | None = None, rank: int | None = None, world_size: int | None = None):
        world = world_size if world_size is not None else get_world_size()
        rank = rank if rank is not None else get_rank()
        ring_size = ring_size if ring_size is not None else world
        assert world % ring_size == 0, f"world size {world} not divisible by ring size {ring_size}"
        self.world_size = world
        self.rank = rank
        self.ring_size = ring_size
        self.ring_index = rank // ring_size          # which sub-ring this rank belongs to
        self.ring_rank = rank % ring_size            # position within the sub-ring
        self.ring_base = self.ring_index * ring_size  # global rank of ring position 0
        # eager sub-ring group creation: new_group is collective, so creating
        # at first topology construction (start of forward, before this
        # layer's other collectives) keeps the creation order deterministic
        # across ranks; init_ring_groups() at startup is the belt-and-braces
        # pattern (only when sizes describe THIS process's world)
        if (world_size is None and rank is None and ring_size < world
                and dist.is_initialized()):
            init_ring_groups(ring_size)

    def global_rank_of(self, ring_rank: int) -> int:
        return self.ring_base + (ring_rank % self.ring_size)

    @property
    def right(self) -> int:
        """Global rank of the next rank around the ring (receives what we send)."""
        return self.global_rank_of(self.ring_rank + 1)

    @property
    def left(self) -> int:
        """Global rank of the previous rank around the ring (sends what we receive)."""
        return self.global_rank_of(self.ring_rank - 1)

    def process_group(self):
        """The process group of THIS rank's ring (None = default/world group).

        Groups are created eagerly at topology construction (or by an explicit
        ``init_ring_groups`` call at startup) and cached per default-group
        identity, so re-initialized process groups never see stale handles."""
        if self.ring_size == self.world_size or not dist.is_initialized():
            return None
        key = (_default_group_id(), self.world_size, self.ring_size)
        if key not in _RING_GROUPS:
            init_ring_groups(self.ring_size)
        return _RING_GROUPS[key][self.ring_index]

    def source_of_hop(self, hop: int) -> int:
        """Ring rank whose original shard this rank holds after ``hop`` passes.

        Shards travel rightward, so after ``hop`` passes rank r holds the
        shard that originated at ring rank ``r - hop``.
        """
        return (self.ring_rank - hop) % self.ring_size
