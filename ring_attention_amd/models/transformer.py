"""RingTransformer — the end-to-end test/bench model.

Capability parity with the reference's RingTransformer
(/root/reference/ring_attention_pytorch/ring_attention.py:468-685): token
embedding, depth x (RingAttention + FeedForward) pre-norm residual stack,
auto label derivation, sequence padding + striped permutation + batch->seq
sharding done ONCE at the model level, shared rank-aware rotary across
layers, per-layer lookback schedule, sharded CE loss, inference re-gather.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import Tensor, nn

from ..parallel import get_world_size, is_distributed
from ..utils.sharding import (
    maybe_pad_seq_and_mask,
    plan_ring_shard,
    sharded_batch_to_sharded_seq,
    sharded_seq_to_sharded_batch,
    stripe_permute,
    stripe_unpermute,
)
from .attention import RMSNorm, RingAttention
from .rotary import RingRotaryEmbedding


class FeedForward(nn.Module):
    def __init__(self, dim: int, mult: int = 4):
        super().__init__()
        inner = int(dim * mult)
        self.net = nn.Sequential(
            RMSNorm(dim),
            nn.Linear(dim, inner, bias=False),
            nn.GELU(),
            nn.Linear(inner, dim, bias=False),
        )

    def forward(self, x: Tensor) -> Tensor:
        return self.net(x)


class RingTransformer(nn.Module):
    def __init__(
        self,
        *,
        num_tokens: int,
        dim: int,
        depth: int,
        causal: bool = False,
        dim_head: int = 64,
        heads: int = 8,
        ff_mult: int = 4,
        num_grouped_query_heads: int = 1,
        bucket_size: int = 512,
        ring_attn: bool = False,
        striped_ring_attn: bool = False,
        ring_seq_size: int = 512,
        auto_shard_seq: bool | None = None,
        max_lookback_seq_len: int | tuple[int | None, ...] | None = None,
        rotary_embed_theta: float = 10000.0,
        ignore_index: int = -1,
        force_regular_attn: bool = False,
        use_hip_kernel: bool | None = None,
        fp8_inference: bool = False,
    ):
        super().__init__()
        self.ring_attn = ring_attn
        self.striped_ring_attn = striped_ring_attn
        self.ring_seq_size = ring_seq_size
        self.bucket_size = bucket_size
        self.ignore_index = ignore_index
        self.auto_shard_seq = auto_shard_seq if auto_shard_seq is not None else ring_attn

        self.token_emb = nn.Embedding(num_tokens, dim)
        self.rotary = RingRotaryEmbedding(
            dim=dim_head, ring=ring_attn, striped=striped_ring_attn,
            theta=rotary_embed_theta)

        # per-layer lookback schedule (int -> same for all layers)
        if max_lookback_seq_len is None or isinstance(max_lookback_seq_len, int):
            lookbacks = (max_lookback_seq_len,) * depth
        else:
            assert len(max_lookback_seq_len) == depth
            lookbacks = tuple(max_lookback_seq_len)

        self.layers = nn.ModuleList()
        for layer_lookback in lookbacks:
            attn = RingAttention(
                dim=dim, dim_head=dim_head, heads=heads,
                num_grouped_query_heads=num_grouped_query_heads,
                causal=causal, bucket_size=bucket_size, ring_attn=ring_attn,
                ring_seq_size=ring_seq_size,
                max_lookback_seq_len=layer_lookback,
                striped_ring_attn=striped_ring_attn,
                auto_shard_seq=False,  # sharding happens once, here at model level
                prenorm=True,
                force_regular_attn=force_regular_attn,
                rotary_embed=False,    # shared rotary passed in per forward
                use_hip_kernel=use_hip_kernel,
                fp8_inference=fp8_inference,
            )
            ff = FeedForward(dim=dim, mult=ff_mult)
            self.layers.append(nn.ModuleList([attn, ff]))

        self.norm = RMSNorm(dim)
        self.to_logits = nn.Linear(dim, num_tokens, bias=False)

    def forward(
        self,
        x: Tensor,                      # (b, n) token ids (batch-sharded across ranks)
        mask: Tensor | None = None,
        labels: Tensor | None = None,
        return_loss: bool | None = None,
        force_ring_reduce_off: bool = False,
        ring_size: int | None = None,
    ) -> Tensor:
        return_loss = return_loss if return_loss is not None else labels is not None
        if return_loss and labels is None:
            x, labels = x[:, :-1], x[:, 1:]

        orig_seq_len = x.shape[1]
        shard_seq = self.ring_attn and self.auto_shard_seq and is_distributed() \
            and not force_ring_reduce_off
        num_sharded_batches = 1
        batch_sizes = None
        ring_size = ring_size if ring_size is not None else get_world_size()

        if shard_seq:
            padded_len, shard, chunks = plan_ring_shard(
                x.shape[1], self.ring_seq_size, self.bucket_size, get_world_size())
            x, mask = maybe_pad_seq_and_mask(x, mask, padded_len)
            if labels is not None:
                pad = x.shape[1] - labels.shape[1]
                if pad:
                    labels = torch.nn.functional.pad(labels, (0, pad),
                                                     value=self.ignore_index)
            if self.striped_ring_attn:
                x = stripe_permute(x, chunks)
                if labels is not None:
                    labels = stripe_permute(labels, chunks)
                if mask is not None:
                    mask = stripe_permute(mask, chunks)
            (x, mask), batch_sizes, num_sharded_batches = \
                sharded_batch_to_sharded_seq(x, mask, shard)
            if labels is not None:
                (labels, _), _, _ = sharded_batch_to_sharded_seq(labels, None, shard)
            ring_size = get_world_size() // num_sharded_batches

        tokens = self.token_emb(x)
        rotary = self.rotary(tokens.shape[1], ring_size)

        for attn, ff in self.layers:
            tokens = attn(tokens, mask=mask, rotary_emb=rotary,
                          force_ring_reduce_off=force_ring_reduce_off,
                          ring_size=ring_size) + tokens
            tokens = ff(tokens) + tokens

        logits = self.to_logits(self.norm(tokens))

        if return_loss:
            # loss stays sharded: each rank's CE over its shard; DDP/optimizer
            # all-reduce handles gradient averaging across ranks
            return F.cross_entropy(
                logits.permute(0, 2, 1), labels, ignore_index=self.ignore_index)

        if shard_seq:
            logits = sharded_seq_to_sharded_batch(logits, batch_sizes, num_sharded_batches)
            if self.striped_ring_attn:
                logits = stripe_unpermute(logits, chunks)
            logits = logits[:, :orig_seq_len]
        return logits
