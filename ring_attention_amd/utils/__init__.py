from .sharding import (
    maybe_pad_seq_and_mask,
    pad_at_dim,
    pad_to_multiple,
    plan_ring_shard,
    sharded_batch_to_sharded_seq,
    sharded_seq_to_sharded_batch,
    stripe_permute,
    stripe_unpermute,
)

__all__ = [
    "maybe_pad_seq_and_mask", "pad_at_dim", "pad_to_multiple", "plan_ring_shard",
    "sharded_batch_to_sharded_seq", "sharded_seq_to_sharded_batch",
    "stripe_permute", "stripe_unpermute",
]
