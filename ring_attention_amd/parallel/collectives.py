"""Autograd-aware collectives over torch.distributed (RCCL on ROCm, gloo on CPU).

Capability parity with the reference's variable-dim AllGather / split_by_rank
(/root/reference/ring_attention_pytorch/distributed.py:43-127), built
MI355X-first: on an 8-GPU node RCCL stripes an all-gather across all 7 xGMI
links per GPU, so batch<->sequence resharding goes through plain collectives
rather than ring passes.
"""

from __future__ import annotations

import torch
import torch.distributed as dist
from torch import Tensor
from torch.autograd import Function

from .topology import get_rank, get_world_size, is_distributed


def all_gather_same_dim(t: Tensor) -> list[Tensor]:
    world = get_world_size()
    t = t.contiguous()
    out = [torch.empty_like(t) for _ in range(world)]
    dist.all_gather(out, t)
    return out


def gather_sizes(t: Tensor, dim: int) -> Tensor:
    """All-gather the size of ``t`` along ``dim`` from every rank -> 1-D int64."""
    size = torch.tensor([t.shape[dim]], device=t.device, dtype=torch.long)
    return torch.cat(all_gather_same_dim(size), dim=0)


def all_gather_variable_dim(t: Tensor, dim: int = 0, sizes: Tensor | None = None) -> tuple[Tensor, Tensor]:
    """All-gather tensors whose ``dim`` extent may differ per rank.

    Pads to the max extent, gathers once (one large RCCL call beats many
    small ones on xGMI), then slices the padding back out.
    Returns (gathered, sizes).
    """
    if not is_distributed():
        sizes = torch.tensor([t.shape[dim]], device=t.device, dtype=torch.long)
        return t, sizes

    if sizes is None:
        sizes = gather_sizes(t, dim)

    max_size = int(sizes.amax().item())
    if bool((sizes == max_size).all()):
        gathered = torch.cat(all_gather_same_dim(t), dim=dim)
        return gathered, sizes

    pad = max_size - t.shape[dim]
    padded = torch.nn.functional.pad(
        t, _pad_spec(t.ndim, dim, pad)
    )
    chunks = all_gather_same_dim(padded)
    trimmed = [c.narrow(dim, 0, int(s.item())) for c, s in zip(chunks, sizes)]
    gathered = torch.cat(trimmed, dim=dim)
    return gathered, sizes


def _pad_spec(ndim: int, dim: int, amount: int) -> list[int]:
    # F.pad takes pairs from the LAST dim backwards
    dim = dim % ndim
    spec = [0, 0] * (ndim - dim)
    spec[2 * (ndim - dim) - 1] = amount
    return spec


class AllGatherFunction(Function):
    """Autograd all-gather along ``dim``.

    Backward is the mathematically correct distributed adjoint: a
    reduce-scatter (sum every rank's gradient for the gathered tensor, keep
    this rank's slice).  The reference instead took the LOCAL grad's own
    slice with no reduction (distributed.py:103-107), dropping every other
    rank's contribution to this rank's input — only tolerable under DDP
    parameter averaging, wrong for input gradients.
    """

    @staticmethod
    def forward(ctx, x: Tensor, dim: int, sizes: Tensor | None):
        is_bool = x.dtype == torch.bool
        if is_bool:
            x = x.int()
        x, sizes = all_gather_variable_dim(x, dim=dim, sizes=sizes)
        if is_bool:
            x = x.bool()
        ctx.dim = dim
        ctx.sizes = sizes
        return x, sizes

    @staticmethod
    def backward(ctx, grad: Tensor, _grad_sizes):
        rank = get_rank()
        splits = ctx.sizes.tolist()
        if is_distributed():
            grad = grad.contiguous()
            dist.all_reduce(grad)   # reduce-scatter expressed as all-reduce + slice
        grads = grad.split(splits, dim=ctx.dim)
        return grads[rank].contiguous(), None, None


class AllGather(torch.nn.Module):
    def __init__(self, dim: int = 0):
        super().__init__()
        self.dim = dim

    def forward(self, x: Tensor, sizes: Tensor | None = None) -> tuple[Tensor, Tensor]:
        return AllGatherFunction.apply(x, self.dim, sizes)


def all_gather(x: Tensor, dim: int = 0, sizes: Tensor | None = None) -> tuple[Tensor, Tensor]:
    return AllGatherFunction.apply(x, dim, sizes)


def split_by_rank(xs: list[Tensor]) -> Tensor:
    """Each rank keeps its own element of a per-rank list."""
    if not is_distributed():
        assert len(xs) == 1
        return xs[0]
    return xs[get_rank()]


def gather_cat(t: Tensor, dim: int, group=None) -> Tensor:
    """Plain (non-autograd) all-gather concatenated along ``dim`` in rank order.

    ``group`` restricts the gather to a sub-ring's process group."""
    if not is_distributed():
        return t
    world = dist.get_world_size(group) if group is not None else get_world_size()
    t = t.contiguous()
    if dist.get_backend() == "nccl":
        out = torch.empty((world,) + tuple(t.shape), device=t.device, dtype=t.dtype)
        dist.all_gather_into_tensor(out.view(world, -1), t.view(-1), group=group)
        chunks = list(out.unbind(0))
    else:
        chunks = [torch.empty_like(t) for _ in range(world)]
        dist.all_gather(chunks, t, group=group)
    return torch.cat(chunks, dim=dim)


def reduce_scatter_chunks(chunks: Tensor, group=None) -> Tensor:
    """``chunks`` (W, ...) — sum chunk r across ranks, return this rank's chunk.

    RCCL reduce-scatter stripes across every xGMI link; the gloo fallback is
    all-reduce + slice.  ``group`` restricts to a sub-ring."""
    if not is_distributed():
        assert chunks.shape[0] == 1
        return chunks[0]
    world = dist.get_world_size(group) if group is not None else get_world_size()
    my = dist.get_rank(group) if group is not None else get_rank()
    assert chunks.shape[0] == world
    chunks = chunks.contiguous()
    if dist.get_backend() == "nccl":
        out = torch.empty_like(chunks[0])
        dist.reduce_scatter_tensor(out.view(-1), chunks.view(world, -1).reshape(-1),
                                   group=group)
        return out
    dist.all_reduce(chunks, group=group)
    return chunks[my].clone()
