"""Rank-aware rotary embeddings.

Capability parity with the reference's RingRotaryEmbedding
(/root/reference/ring_attention_pytorch/ring_attention.py:102-172): each rank
generates the rotary phases for ITS OWN global token positions, so no
position tensor ever crosses the wire.

Position maps (shard length n, ring size R, ring rank r):
- plain ring (contiguous layout):  r * n + arange(n)
- striped ring (stride-R layout):  arange(n) * R + r
All trig in fp32 regardless of activation dtype.
"""

from __future__ import annotations

import torch
from torch import Tensor, nn

from ..parallel import RingTopology, is_distributed


class RingRotaryEmbedding(nn.Module):
    def __init__(
        self,
        dim: int,
        ring: bool = False,
        striped: bool = False,
        theta: float = 10000.0,
    ):
        super().__init__()
        self.dim = dim
        self.ring = ring
        self.striped = striped
        inv_freq = 1.0 / (theta ** (torch.arange(0, dim, 2).float() / dim))
        self.register_buffer("inv_freq", inv_freq, persistent=False)

    @torch.autocast("cuda", enabled=False)
    def forward(self, seq_len: int, ring_size: int | None = None) -> Tensor:
        """Returns freqs (seq_len, dim) for this rank's shard positions."""
        device = self.inv_freq.device
        if self.ring and is_distributed():
            topo = RingTopology(ring_size)
            if self.striped:
                pos = torch.arange(seq_len, device=device) * topo.ring_size + topo.ring_rank
            else:
                pos = topo.ring_rank * seq_len + torch.arange(seq_len, device=device)
        else:
            pos = torch.arange(seq_len, device=device)
        freqs = torch.einsum("i,j->ij", pos.float(), self.inv_freq.float())
        return torch.cat((freqs, freqs), dim=-1)  # (n, dim)


def rotate_half(x: Tensor) -> Tensor:
    x1, x2 = x.chunk(2, dim=-1)
    return torch.cat((-x2, x1), dim=-1)


class _FusedRotary(torch.autograd.Function):
    """One-kernel rotary on GPU (csrc/rotary.hip).  The rotation is
    orthogonal, so backward is the same kernel with sin negated."""

    @staticmethod
    def forward(ctx, t, cos_t, sin_t):
        from ..ops import hip_ext
        out = hip_ext.require().rotary_apply(t.contiguous(), cos_t, sin_t, 1.0)
        ctx.save_for_backward(cos_t, sin_t)
        return out

    @staticmethod
    def backward(ctx, grad):
        from ..ops import hip_ext
        cos_t, sin_t = ctx.saved_tensors
        g = hip_ext.require().rotary_apply(
            grad.to(torch.bfloat16).contiguous(), cos_t, sin_t, -1.0)
        return g, None, None


@torch.autocast("cuda", enabled=False)
def apply_rotary_pos_emb(freqs: Tensor, t: Tensor) -> Tensor:
    """freqs (n, d); t (b, n, h, d) -> rotated t (same dtype as input)."""
    d = t.shape[-1]
    if (t.is_cuda and t.dtype == torch.bfloat16 and d in (64, 128)
            and t.dim() == 4):
        from ..ops import hip_ext
        if hip_ext.available():
            half = freqs[:, :d // 2].float()
            return _FusedRotary.apply(t, half.cos().contiguous(),
                                      half.sin().contiguous())
    dtype = t.dtype
    t = t.float()
    f = freqs[None, :, None, :]
    out = t * f.cos() + rotate_half(t) * f.sin()
    return out.to(dtype)
