"""RingAttention module — the attention layer of the framework.

Capability parity with the reference's RingAttention
(/root/reference/ring_attention_pytorch/ring_attention.py:283-466): RMSNorm
prenorm fused into the qkv projection, grouped-query heads (only kv heads
circulate the ring), rank-aware rotary, optional auto sequence sharding with
striped permutation, per-layer lookback, and dispatch between the eager
oracle, the portable bucketed ring function, and the CDNA4 HIP kernel path.

On a GPU the HIP kernel path is THE path: if the extension is missing the
module raises rather than silently falling back to a slow implementation.
"""

from __future__ import annotations

import torch
from torch import Tensor, nn

from ..ops.reference import default_attention
from ..ops.ring_flash import ring_flash_attn_
from ..parallel import get_world_size, is_distributed
from ..utils.sharding import (
    maybe_pad_seq_and_mask,
    plan_ring_shard,
    sharded_batch_to_sharded_seq,
    sharded_seq_to_sharded_batch,
    stripe_permute,
    stripe_unpermute,
)
from .rotary import RingRotaryEmbedding, apply_rotary_pos_emb


class RMSNorm(nn.Module):
    """x / rms(x) * gamma — via torch's fused rms_norm kernel (one pass;
    the reference composed normalize + 2 muls, ring_attention.py:470-477)."""

    def __init__(self, dim: int):
        super().__init__()
        self.dim = dim
        self.gamma = nn.Parameter(torch.ones(dim))

    def forward(self, x: Tensor) -> Tensor:
        return torch.nn.functional.rms_norm(x, (self.dim,), self.gamma, eps=1e-24)


class RingAttention(nn.Module):
    def __init__(
        self,
        dim: int,
        *,
        dim_head: int = 64,
        heads: int = 8,
        num_grouped_query_heads: int = 1,
        causal: bool = False,
        bucket_size: int = 512,
        ring_attn: bool = False,
        ring_seq_size: int = 512,       # per-rank shard size (reference's naming kept)
        max_lookback_seq_len: int | None = None,
        striped_ring_attn: bool = False,
        auto_shard_seq: bool | None = None,
        prenorm: bool = True,
        force_regular_attn: bool = False,
        rotary_embed: bool = False,
        rotary_embed_theta: float = 10000.0,
        softclamp_qk_sim: bool = False,
        softclamp_value: float = 50.0,
        use_hip_kernel: bool | None = None,
        fp8_inference: bool = False,
    ):
        super().__init__()
        assert heads % num_grouped_query_heads == 0, (
            f"query heads ({heads}) must be divisible by groups ({num_grouped_query_heads})")
        assert (not ring_attn) or ring_seq_size % bucket_size == 0

        kv_heads = heads // num_grouped_query_heads
        self.heads = heads
        self.kv_heads = kv_heads
        self.num_grouped_query_heads = num_grouped_query_heads
        self.dim_head = dim_head
        self.causal = causal
        self.bucket_size = bucket_size
        self.ring_attn = ring_attn
        self.ring_seq_size = ring_seq_size
        self.max_lookback_seq_len = max_lookback_seq_len
        self.striped_ring_attn = striped_ring_attn
        self.force_regular_attn = force_regular_attn
        self.softclamp_qk_sim = softclamp_qk_sim
        self.softclamp_value = softclamp_value
        self.use_hip_kernel = use_hip_kernel  # None = auto (on GPU)
        # MX-FP8 inference: no-grad forwards route to the e4m3 serving path
        # (ops/fp8.py) when its scope applies — non-striped, no mask/bias/
        # softclamp/lookback.  Training steps and out-of-scope calls keep
        # the bf16 path automatically.
        self.fp8_inference = fp8_inference

        assert not (striped_ring_attn and not causal), "striped ring attention requires causal"
        self.auto_shard_seq = auto_shard_seq if auto_shard_seq is not None else ring_attn
        assert not (self.auto_shard_seq and not ring_attn)

        self.rotary_embed = None
        if rotary_embed:
            self.rotary_embed = RingRotaryEmbedding(
                dim=dim_head, ring=ring_attn, striped=striped_ring_attn,
                theta=rotary_embed_theta)

        dim_q = dim_head * heads
        dim_kv = dim_head * kv_heads
        self.qkv_split = (dim_q, dim_kv, dim_kv)
        self.to_qkv = nn.Sequential(
            RMSNorm(dim) if prenorm else nn.Identity(),
            nn.Linear(dim, dim_q + 2 * dim_kv, bias=False),
        )
        self.to_out = nn.Linear(dim_q, dim, bias=False)

    def forward(
        self,
        x: Tensor,
        mask: Tensor | None = None,
        rotary_emb: Tensor | None = None,
        force_ring_reduce_off: bool = False,
        ring_size: int | None = None,
    ) -> Tensor:
        ring_size = ring_size if ring_size is not None else get_world_size()
        ring_attn = self.ring_attn and is_distributed()
        auto_shard_seq = self.auto_shard_seq and is_distributed()

        orig_seq_len = x.shape[1]
        num_sharded_batches = 1
        batch_sizes = None
        if auto_shard_seq:
            padded_len, shard, chunks = plan_ring_shard(
                x.shape[1], self.ring_seq_size, self.bucket_size, get_world_size())
            x, mask = maybe_pad_seq_and_mask(x, mask, padded_len)
            if self.striped_ring_attn:
                x = stripe_permute(x, chunks)
                if mask is not None:
                    mask = stripe_permute(mask, chunks, dim=1)
            (x, mask), batch_sizes, num_sharded_batches = \
                sharded_batch_to_sharded_seq(x, mask, shard)
            ring_size = get_world_size() // num_sharded_batches

        b, n, _ = x.shape
        qkv = self.to_qkv(x)
        q, k, v = qkv.split(self.qkv_split, dim=-1)
        q = q.view(b, n, self.heads, self.dim_head)
        k = k.view(b, n, self.kv_heads, self.dim_head)
        v = v.view(b, n, self.kv_heads, self.dim_head)

        if rotary_emb is None and self.rotary_embed is not None:
            rotary_emb = self.rotary_embed(n, ring_size if ring_attn else None)
        if rotary_emb is not None:
            q = apply_rotary_pos_emb(rotary_emb, q)
            k = apply_rotary_pos_emb(rotary_emb, k)

        if self.force_regular_attn:
            out = default_attention(
                q, k, v, mask=mask, causal=self.causal,
                softclamp_qk_sim=self.softclamp_qk_sim,
                softclamp_value=self.softclamp_value)
        else:
            out = self._flash(q, k, v, mask,
                              ring_reduce=ring_attn and not force_ring_reduce_off,
                              ring_size=ring_size)

        out = out.reshape(b, n, self.heads * self.dim_head)
        out = self.to_out(out)

        if auto_shard_seq:
            out = sharded_seq_to_sharded_batch(out, batch_sizes, num_sharded_batches)
            if self.striped_ring_attn:
                out = stripe_unpermute(out, chunks)
            out = out[:, :orig_seq_len]
        return out

    def _flash(self, q, k, v, mask, ring_reduce: bool, ring_size: int) -> Tensor:
        if (self.fp8_inference and not torch.is_grad_enabled()
                and mask is None and not self.striped_ring_attn
                and not self.softclamp_qk_sim
                and self.max_lookback_seq_len is None):
            from ..ops.fp8 import flash_attn_fp8, ring_flash_attn_fp8
            if ring_reduce:
                out, _ = ring_flash_attn_fp8(
                    q, k, v, ring_size=ring_size, causal=self.causal)
            else:
                out, _ = flash_attn_fp8(q, k, v, causal=self.causal)
            return out
        use_hip = self.use_hip_kernel if self.use_hip_kernel is not None else q.is_cuda
        if use_hip and q.is_cuda:
            from ..ops.ring_flash_hip import ring_flash_attn_hip_
            out, _ = ring_flash_attn_hip_(
                q, k, v, mask=mask, causal=self.causal,
                bucket_size=self.bucket_size,
                ring_reduce_col=ring_reduce,
                striped_ring_attn=self.striped_ring_attn,
                max_lookback_seq_len=self.max_lookback_seq_len,
                ring_size=ring_size,
                softclamp_qk_sim=self.softclamp_qk_sim,
                softclamp_value=self.softclamp_value)
            return out
        out, _ = ring_flash_attn_(
            q, k, v, mask=mask, causal=self.causal,
            bucket_size=self.bucket_size,
            ring_reduce_col=ring_reduce,
            striped_ring_attn=self.striped_ring_attn,
            max_lookback_seq_len=self.max_lookback_seq_len,
            ring_size=ring_size,
            softclamp_qk_sim=self.softclamp_qk_sim,
            softclamp_value=self.softclamp_value)
        return out
