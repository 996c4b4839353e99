"""Sequence-sharded data preparation (an item the reference left as an open
TODO — README.md:120 'craft a special Dataset that shards across sequence
length (take into account labels for cross entropy loss)').

``shard_sequence_batch`` performs the SAME plan the model's auto-shard path
computes (derive labels, pad, stripe, split) but statically and without any
collectives — every rank slices its own shard from the host batch, so no
token ever crosses the wire.  Feed the result to a ``RingTransformer`` built
with ``auto_shard_seq=False``.
"""

from __future__ import annotations

from typing import NamedTuple

import torch
from torch import Tensor
from torch.utils.data import Dataset

from .sharding import plan_ring_shard, stripe_permute


class ShardedBatch(NamedTuple):
    ids: Tensor              # (b, shard) this rank's token shard
    labels: Tensor           # (b, shard) aligned next-token labels
    mask: Tensor | None      # (b, shard) key-padding mask (None if no padding)
    ring_size: int
    shard_size: int


def shard_sequence_batch(
    ids: Tensor,                  # (b, n) full token ids (host-side)
    *,
    ring_seq_size: int,
    bucket_size: int,
    world: int,
    rank: int,
    striped: bool = False,
    ignore_index: int = -1,
    labels: Tensor | None = None,
) -> ShardedBatch:
    """Statically derive rank `rank`'s sequence shard of a full batch."""
    if labels is None:
        ids, labels = ids[:, :-1], ids[:, 1:]
    b, n = ids.shape

    padded, shard, chunks = plan_ring_shard(n, ring_seq_size, bucket_size, world)
    assert chunks == world, (
        "pre-sharded data requires one chunk per rank (sequence of "
        f"{n} tokens -> {chunks} chunks for world {world}); lower "
        "ring_seq_size or use the model's auto-shard path for sub-rings")
    pad = padded - n
    mask = None
    if pad:
        mask = torch.ones(b, padded, dtype=torch.bool)
        mask[:, n:] = False
        ids = torch.nn.functional.pad(ids, (0, pad))
        labels = torch.nn.functional.pad(labels, (0, pad), value=ignore_index)
    if striped:
        ids = stripe_permute(ids, chunks)
        labels = stripe_permute(labels, chunks)
        if mask is not None:
            mask = stripe_permute(mask, chunks)

    sl = slice(rank * shard, (rank + 1) * shard)
    return ShardedBatch(ids[:, sl].contiguous(), labels[:, sl].contiguous(),
                        mask[:, sl].contiguous() if mask is not None else None,
                        world, shard)


class ShardedSequenceDataset(Dataset):
    """Wraps a dataset of full token sequences; __getitem__ returns THIS
    rank's shard only, labels aligned, ready for a non-auto-shard
    RingTransformer.  All ranks must iterate in the same order."""

    def __init__(self, sequences: Dataset, *, ring_seq_size: int, bucket_size: int,
                 world: int, rank: int, striped: bool = False, ignore_index: int = -1):
        self.sequences = sequences
        self.kw = dict(ring_seq_size=ring_seq_size, bucket_size=bucket_size,
                       world=world, rank=rank, striped=striped,
                       ignore_index=ignore_index)

    def __len__(self):
        return len(self.sequences)

    def __getitem__(self, idx):
        seq = self.sequences[idx]
        if seq.dim() == 1:
            seq = seq[None]
        return shard_sequence_batch(seq, **self.kw)
