"""Distributed ring flash attention ≡ replicated eager oracle, per-shard, tight tol.

Every rank builds the SAME full (q,k,v) from a shared seed, runs the ring
function on its shard, and compares output AND per-shard dq/dk/dv against the
replicated eager ground truth at fp32 tolerance — the check the reference
lacked (SURVEY.md §4 'gap to learn from').
"""

import pytest
import torch

from ring_attention_amd.ops import default_attention
from ring_attention_amd.ops.ring_flash import ring_flash_attn_

from .distributed_utils import run_distributed


def _striped_positions(n_shard, world, rank):
    # local i on rank r <-> global i * world + r
    return torch.arange(n_shard) * world + rank


def _ring_case(rank, world, causal, striped, groups, mask_on, bucket_size, lookback):
    torch.manual_seed(42)
    b, n_total, h, d = 2, 32 * world, 4, 16
    hk = h // groups
    n = n_total // world
    q = torch.randn(b, n_total, h, d)
    k = torch.randn(b, n_total, hk, d)
    v = torch.randn(b, n_total, hk, d)
    mask = None
    if mask_on:
        mask = torch.rand(b, n_total) > 0.2
        mask[:, :2] = True

    if striped:
        # global g lives at (rank g % world, local g // world)
        perm = torch.arange(n_total).view(-1, world).t().reshape(-1)  # rank-major order
        # rank r's shard = positions r, r+W, ... => indices perm[r*n:(r+1)*n]
        shard_idx = torch.arange(n) * world + rank
    else:
        shard_idx = torch.arange(rank * n, (rank + 1) * n)

    qs = q[:, shard_idx].detach().requires_grad_(True)
    ks = k[:, shard_idx].detach().requires_grad_(True)
    vs = v[:, shard_idx].detach().requires_grad_(True)
    ms = mask[:, shard_idx] if mask is not None else None

    out, _ = ring_flash_attn_(
        qs, ks, vs, mask=ms, causal=causal, bucket_size=bucket_size,
        ring_reduce_col=True, striped_ring_attn=striped,
        max_lookback_seq_len=lookback, ring_size=world,
    )

    # replicated ground truth on the full sequence
    q2 = q.detach().requires_grad_(True)
    k2 = k.detach().requires_grad_(True)
    v2 = v.detach().requires_grad_(True)
    lookback_mask = None
    if lookback is not None:
        # exact token-level sliding window: distance > lookback masked
        pos_ = torch.arange(n_total)
        lookback_mask = (pos_[:, None] - pos_[None, :]) > lookback
    ref = default_attention(q2, k2, v2, mask=mask, causal=causal)
    if lookback is not None:
        # redo with lookback folded into sim via positions trick: do it manually
        from ring_attention_amd.ops.reference import MASK_VALUE
        import torch.nn.functional as F
        scale = d ** -0.5
        kk = k2.repeat(1, 1, groups, 1) if groups > 1 else k2   # reference tile GQA
        vv = v2.repeat(1, 1, groups, 1) if groups > 1 else v2
        sim = torch.einsum("bihd,bjhd->bhij", q2.float(), kk.float()) * scale
        if mask is not None:
            sim = sim.masked_fill(~mask[:, None, None, :], MASK_VALUE)
        pos = torch.arange(n_total)
        cm = pos[None, :] > pos[:, None]
        sim = sim.masked_fill(cm[None, None], MASK_VALUE)
        sim = sim.masked_fill(lookback_mask[None, None], MASK_VALUE)
        ref = torch.einsum("bhij,bjhd->bihd", sim.softmax(-1), vv.float()).to(q2.dtype)

    g = torch.randn(b, n_total, h, d)
    out.backward(g[:, shard_idx])
    ref.backward(g)

    out_err = (out - ref[:, shard_idx]).abs().max().item()
    dq_err = (qs.grad - q2.grad[:, shard_idx]).abs().max().item()
    dk_err = (ks.grad - k2.grad[:, shard_idx]).abs().max().item()
    dv_err = (vs.grad - v2.grad[:, shard_idx]).abs().max().item()
    assert out_err < 1e-5, f"out {out_err}"
    assert dq_err < 1e-5, f"dq {dq_err}"
    assert dk_err < 1e-5, f"dk {dk_err}"
    assert dv_err < 1e-5, f"dv {dv_err}"
    return (out_err, dq_err, dk_err, dv_err)


@pytest.mark.parametrize("world", [2])
@pytest.mark.parametrize("causal,striped", [(False, False), (True, False), (True, True)])
def test_ring_flash_world2(world, causal, striped):
    run_distributed(world, _ring_case, causal, striped, 1, False, 16, None)


def test_ring_flash_world2_mask_gqa():
    run_distributed(2, _ring_case, True, False, 2, True, 16, None)


def test_ring_flash_world2_striped_gqa_mask():
    run_distributed(2, _ring_case, True, True, 2, True, 16, None)


def test_ring_flash_world4_causal():
    run_distributed(4, _ring_case, True, False, 1, False, 16, None)


def test_ring_flash_world4_striped():
    run_distributed(4, _ring_case, True, True, 1, False, 8, None)


def test_ring_flash_world4_lookback():
    # lookback 32 tokens, bucket 16, shard 32 => hops truncated to < 4
    run_distributed(4, _ring_case, True, False, 1, False, 16, 32)


def test_ring_flash_world2_noncausal_mask():
    run_distributed(2, _ring_case, False, False, 1, True, 16, None)
