"""Sequence padding / batch<->sequence resharding / striped permutation.

Capability parity with the reference's shard helpers
(/root/reference/ring_attention_pytorch/ring_attention.py:176-279): pad the
sequence to a multiple of the per-rank shard size, all-gather the batch
(one large RCCL all-gather striping across all xGMI links), fold surplus
batch groups into independent sub-rings, and split the sequence by rank.

Striped layout (this framework's convention, see ops/ring_flash.py): global
position g lives at (ring rank g % R, local index g // R) — the stride-R
interleave that keeps every rank equally busy under causal masking.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import Tensor

from ..parallel import AllGather, get_world_size, split_by_rank


def pad_at_dim(t: Tensor, pad: tuple[int, int], dim: int = -1, value: float = 0.0) -> Tensor:
    dims_from_right = (-dim - 1) if dim < 0 else (t.ndim - dim - 1)
    zeros = (0, 0) * dims_from_right
    return F.pad(t, (*zeros, *pad), value=value)


def pad_to_multiple(x: Tensor, length: int, dim: int = 1, pad_value: float = 0.0) -> tuple[Tensor, int]:
    n = x.shape[dim]
    remainder = n % length
    if remainder == 0:
        return x, 0
    pad_len = length - remainder
    return pad_at_dim(x, (0, pad_len), dim=dim, value=pad_value), pad_len


def plan_ring_shard(seq_len: int, shard_size: int, bucket_size: int, world: int) -> tuple[int, int, int]:
    """Choose (padded_len, shard, chunks): chunks divides world, shard is a
    multiple of bucket_size and >= the requested shard_size when possible,
    chunks * shard >= seq_len with minimal padding.  The shard GROWS past the
    requested size when the sequence exceeds world * shard_size (the
    reference simply asserted in that case)."""
    import math
    best = None
    for chunks in range(1, world + 1):
        if world % chunks != 0:
            continue
        shard = max(shard_size, math.ceil(seq_len / chunks / bucket_size) * bucket_size)
        padded = chunks * shard
        key = (padded, -chunks)       # least padding, then most parallelism
        if best is None or key < best[0]:
            best = (key, padded, shard, chunks)
    _, padded, shard, chunks = best
    return padded, shard, chunks


def maybe_pad_seq_and_mask(
    x: Tensor, mask: Tensor | None, target_len: int,
) -> tuple[Tensor, Tensor | None]:
    """Pad the sequence (dim 1) up to ``target_len``; synthesize/extend mask."""
    shape = x.shape[:2]
    pad_len = target_len - x.shape[1]
    assert pad_len >= 0
    if pad_len == 0:
        return x, mask
    x = pad_at_dim(x, (0, pad_len), dim=1)
    if mask is None:
        mask = torch.ones(shape, device=x.device, dtype=torch.bool)
    mask = pad_at_dim(mask, (0, x.shape[1] - mask.shape[1]), dim=1, value=False)
    return x, mask


def stripe_permute(x: Tensor, ring_size: int, dim: int = 1) -> Tensor:
    """Contiguous -> striped: output chunk r = global positions {r, r+R, ...}."""
    n = x.shape[dim]
    assert n % ring_size == 0
    shard = n // ring_size
    idx = (torch.arange(shard, device=x.device)[None, :] * ring_size
           + torch.arange(ring_size, device=x.device)[:, None]).reshape(-1)
    return x.index_select(dim, idx)


def stripe_unpermute(x: Tensor, ring_size: int, dim: int = 1) -> Tensor:
    """Inverse of stripe_permute."""
    n = x.shape[dim]
    assert n % ring_size == 0
    shard = n // ring_size
    idx = (torch.arange(ring_size, device=x.device)[None, :] * shard
           + torch.arange(shard, device=x.device)[:, None]).reshape(-1)
    return x.index_select(dim, idx)


def sharded_batch_to_sharded_seq(
    x: Tensor, mask: Tensor | None, shard_size: int
) -> tuple[tuple[Tensor, Tensor | None], Tensor, int]:
    """Each rank arrives with a batch shard of full sequences (len a multiple of
    ``shard_size``); leaves with its sequence shard.  When the world is larger
    than the number of sequence chunks, surplus ranks carry extra batch groups
    (independent sub-rings).  Returns ((x, mask), batch_sizes, num_sharded_batches)."""
    gather_batch = AllGather(dim=0)
    x, sizes = gather_batch(x)
    if mask is not None:
        mask, _ = gather_batch(mask)

    world = get_world_size()
    chunks_per_seq = x.shape[1] // shard_size
    assert world % chunks_per_seq == 0, (
        f"world {world} not divisible by per-sequence chunk count {chunks_per_seq}")
    num_sharded_batches = world // chunks_per_seq

    # fold batch groups into the sequence dim: (b s) n ... -> b (s n) ...
    b_total = x.shape[0]
    assert b_total % num_sharded_batches == 0
    b = b_total // num_sharded_batches
    x = x.reshape(b, num_sharded_batches * x.shape[1], *x.shape[2:])
    xs = x.split(shard_size, dim=1)
    x = split_by_rank(list(xs))

    if mask is not None:
        mask = mask.reshape(b, num_sharded_batches * mask.shape[1])
        mask = split_by_rank(list(mask.split(shard_size, dim=-1)))

    return (x, mask), sizes, num_sharded_batches


def sharded_seq_to_sharded_batch(logits: Tensor, sizes: Tensor, num_sharded_batches: int = 1) -> Tensor:
    """Inverse: gather sequence shards, unfold batch groups, split batch by rank."""
    gather_seq = AllGather(dim=-2)
    logits, _ = gather_seq(logits)
    b = logits.shape[0]
    logits = logits.reshape(b * num_sharded_batches,
                            logits.shape[1] // num_sharded_batches, *logits.shape[2:])
    logits = split_by_rank(list(logits.split(sizes.tolist(), dim=0)))
    return logits
