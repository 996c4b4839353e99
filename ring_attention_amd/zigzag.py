"""Zig-zag context parallelism (the Llama-3 scheme, arXiv:2407.21783).

Capability parity with the reference's zig_zag_attention
(/root/reference/ring_attention_pytorch/zig_zag_attention.py:35-140): pad the
sequence to a multiple of 2*world; rank r owns chunks r and 2W-1-r (causal
load balance); K/V are all-gathered over the sequence (one RCCL all-gather
that stripes across all 7 xGMI links — bandwidth-optimal on one MI355X node,
unlike a ring which is single-link-bound) and attention runs with an explicit
mask derived from the exported positions.

Layout parity with the reference: (batch, heads, seq, dim) — "b h n d".
"""

from __future__ import annotations

import math
from collections import namedtuple

import torch
import torch.nn.functional as F
from torch import Tensor

from .ops.reference import MASK_VALUE
from .parallel import AllGather, get_rank, get_world_size

ShardOutput = namedtuple("ShardOutput", [
    "local_sequence",
    "query_positions",
    "key_value_positions",
])


def zig_zag_pad_seq(t: Tensor):
    """Pad dim -2 to a multiple of 2*world; returns (padded, inverse_fn)."""
    seq_len = t.shape[-2]
    chunks = 2 * get_world_size()
    padded = math.ceil(seq_len / chunks) * chunks
    t = F.pad(t, (0, 0, 0, padded - seq_len), value=0.0)

    def inverse(out: Tensor) -> Tensor:
        return out[..., :seq_len, :]

    return t, inverse


def zig_zag_shard(t: Tensor, all_gather_batch: bool = False):
    """Rank r keeps chunks (r, 2W-1-r); exports global positions for rotary/masking.

    Returns (ShardOutput(local_sequence, query_positions, key_value_positions),
    inverse_fn)."""
    device, seq_len = t.device, t.shape[-2]
    rank, world = get_rank(), get_world_size()

    gather_sizes = None
    if all_gather_batch:
        t, gather_sizes = AllGather(dim=0)(t)

    chunks = 2 * world
    chunk_size = seq_len // chunks
    pieces = t.chunk(chunks, dim=-2)
    local = torch.cat((pieces[rank], pieces[chunks - 1 - rank]), dim=-2).contiguous()

    pos = torch.arange(seq_len, device=device).view(chunks, chunk_size)
    first, second = pos.chunk(2, dim=0)
    paired = torch.stack((first, second.flip(dims=(0,))), dim=1)  # (world, 2, chunk)

    q_indices = paired[rank].reshape(-1)
    kv_indices = paired.reshape(-1)

    def inverse(two_chunks: Tensor) -> Tensor:
        tc = two_chunks.reshape(*two_chunks.shape[:-2], 2,
                                two_chunks.shape[-2] // 2, two_chunks.shape[-1])
        all_chunks, _ = AllGather(dim=-3)(tc)               # (b, 2W, chunk, d) pair-ordered
        shape = all_chunks.shape
        all_chunks = all_chunks.reshape(*shape[:-3], shape[-3] // 2, 2, *shape[-2:])
        first_half = all_chunks[..., 0, :, :]
        second_half = all_chunks[..., 1, :, :].flip(dims=(-3,))
        out = torch.cat((first_half, second_half), dim=-3)
        out = out.reshape(*out.shape[:-3], -1, out.shape[-1])
        if all_gather_batch:
            out = out.split(gather_sizes.tolist(), dim=0)
            out = out[rank]
        return out

    return ShardOutput(local, q_indices, kv_indices), inverse


def zig_zag_attn(
    q: Tensor,                      # (b, h, i, d)
    k: Tensor,                      # (b, hk, j, d)  local shard
    v: Tensor,                      # (b, hk, j, dv)
    dropout: float = 0.0,
    attn_mask: Tensor | None = None,  # bool, True = attend
    causal: bool = False,
    q_chunk_starts: tuple[int, int] | None = None,
    kv_valid_len: int | None = None,   # unpadded global length (pad keys masked)
) -> Tensor:
    """Zig-zag attention over all-gathered KV.

    Fast path (GPU): when ``causal`` and ``q_chunk_starts`` (the global start
    positions of this rank's two chunks, derivable from zig_zag_shard's
    query_positions) are given, the CDNA4 flash kernel runs once per chunk
    with an offset-causal mask — no O(n*N) mask tensor at all.  Otherwise the
    portable masked path runs (parity with the reference API, which takes an
    explicit attn_mask).
    """
    heads, kv_heads = q.shape[1], k.shape[1]
    assert heads % kv_heads == 0
    groups = heads // kv_heads

    gather_seq = AllGather(dim=-2)
    k, _ = gather_seq(k)
    v, _ = gather_seq(v)

    if (causal and q_chunk_starts is not None and q.is_cuda
            and dropout == 0.0 and attn_mask is None):
        from .ops.ring_flash_hip import flash_attn_offset
        n = q.shape[-2]
        half = n // 2
        # the gather is RANK-major: [r0 chunks (0, 2W-1), r1 chunks (1, 2W-2),
        # ...] — the offset-causal kernel needs KV in GLOBAL position order,
        # so reorder chunks first (differentiable; no-op at world 1)
        world = get_world_size()
        if world > 1:
            bk, hkk, N, dk_ = k.shape
            ch = N // (2 * world)
            r = torch.arange(world, device=k.device)
            idx = torch.empty(2 * world, dtype=torch.long, device=k.device)
            idx[r] = 2 * r                      # global chunk r <- rank r local 0
            idx[2 * world - 1 - r] = 2 * r + 1  # global 2W-1-r <- rank r local 1
            k = k.reshape(bk, hkk, 2 * world, ch, dk_).index_select(2, idx).reshape(bk, hkk, N, dk_)
            v = v.reshape(bk, hkk, 2 * world, ch, v.shape[-1]).index_select(2, idx).reshape(bk, hkk, N, v.shape[-1])
        # to (b, n, h, d) layout for the kernel; pad keys are a global-order
        # suffix, so masking them is a slice (grads zero-pad automatically)
        q_ = q.permute(0, 2, 1, 3)
        k_ = k.permute(0, 2, 1, 3)
        v_ = v.permute(0, 2, 1, 3)
        if kv_valid_len is not None:
            k_ = k_[:, :kv_valid_len]
            v_ = v_[:, :kv_valid_len]
        # GQA: the HIP kernels pair q head qh with kv head qh % hk natively
        # (the reference tile convention, ring_attention.py:86-89) — no head
        # permutation needed
        outs = []
        for c, start in enumerate(q_chunk_starts):
            qc = q_[:, c * half:(c + 1) * half]
            outs.append(flash_attn_offset(qc, k_, v_, q_offset=start, causal=True))
        out = torch.cat(outs, dim=1)
        return out.permute(0, 2, 1, 3)
    if groups > 1:
        # repeat pattern parity with the reference: 'b h n d -> b (g h) n d'
        k = k.repeat(1, groups, 1, 1)
        v = v.repeat(1, groups, 1, 1)

    if causal and attn_mask is None:
        # portable path must honor causal too (same semantics as the fast
        # path), not silently compute full attention: derive the mask from the
        # chunk-start positions.  KV after the gather is RANK-major
        # [r0:(0, 2W-1), r1:(1, 2W-2), ...] — build its global positions.
        if q_chunk_starts is None:
            raise ValueError(
                "zig_zag_attn(causal=True) on the portable path needs "
                "q_chunk_starts (from zig_zag_shard's query_positions) or an "
                "explicit attn_mask encoding causality")
        n, N = q.shape[-2], k.shape[-2]
        half = n // 2
        world = get_world_size()
        ch = N // (2 * world)
        r = torch.arange(world, device=q.device)
        kpos = torch.stack((r * ch, (2 * world - 1 - r) * ch), dim=1).reshape(-1)
        kpos = (kpos[:, None] + torch.arange(ch, device=q.device)[None, :]).reshape(-1)
        qpos = torch.cat([q_chunk_starts[0] + torch.arange(half, device=q.device),
                          q_chunk_starts[1] + torch.arange(n - half, device=q.device)])
        attn_mask = kpos[None, :] <= qpos[:, None]
        if kv_valid_len is not None:
            attn_mask = attn_mask & (kpos[None, :] < kv_valid_len)

    dtype = q.dtype
    scale = q.shape[-1] ** -0.5
    sim = torch.einsum("bhid,bhjd->bhij", q.float(), k.float()) * scale
    if attn_mask is not None:
        sim = sim.masked_fill(~attn_mask, MASK_VALUE)
    attn = sim.softmax(dim=-1)
    if dropout > 0.0:
        attn = F.dropout(attn, p=dropout)
    out = torch.einsum("bhij,bhjd->bhid", attn, v.float())
    return out.to(dtype)
