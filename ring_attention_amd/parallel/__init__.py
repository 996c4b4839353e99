from .topology import RingTopology, get_rank, get_world_size, is_distributed
from .collectives import (
    AllGather,
    AllGatherFunction,
    all_gather,
    all_gather_same_dim,
    all_gather_variable_dim,
    gather_sizes,
    split_by_rank,
)
from .ring_pass import (
    RingAccumulator,
    RingInfo,
    all_ring_pass,
    null_ring_pass,
    one_ring_pass,
    ring_pass,
)

__all__ = [
    "RingTopology", "get_rank", "get_world_size", "is_distributed",
    "AllGather", "AllGatherFunction", "all_gather", "all_gather_same_dim",
    "all_gather_variable_dim", "gather_sizes", "split_by_rank",
    "RingAccumulator", "RingInfo", "all_ring_pass", "null_ring_pass",
    "one_ring_pass", "ring_pass",
]
