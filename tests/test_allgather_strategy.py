"""All-gather-KV strategy ≡ ring strategy (oracle forms, gloo world 2/4).

Output AND per-shard dq/dk/dv must match — the all-gather path's backward is
the autograd composition (reduce-scatter adjoint of the gather + local
attention), which must reproduce the hand-written ring backward exactly.
"""

import pytest
import torch

from ring_attention_amd.ops.ring_flash import (
    ring_flash_attn_,
    ring_flash_attn_allgather_,
)

from .distributed_utils import run_distributed


def _case(rank, world, causal, striped, groups, mask_on, lookback):
    torch.manual_seed(21)
    b, n, h, d = 2, 32, 4, 16
    hk = h // groups
    q = torch.randn(b, n, h, d)
    k = torch.randn(b, n, hk, d)
    v = torch.randn(b, n, hk, d)
    mask = None
    if mask_on:
        mask = torch.rand(b, n) > 0.2
        mask[:, :2] = True

    q1 = q.clone().requires_grad_(True)
    k1 = k.clone().requires_grad_(True)
    v1 = v.clone().requires_grad_(True)
    out1, lse1 = ring_flash_attn_(
        q1, k1, v1, mask=mask, causal=causal, bucket_size=16,
        ring_reduce_col=True, striped_ring_attn=striped,
        max_lookback_seq_len=lookback, ring_size=world)

    q2 = q.clone().requires_grad_(True)
    k2 = k.clone().requires_grad_(True)
    v2 = v.clone().requires_grad_(True)
    out2, lse2 = ring_flash_attn_allgather_(
        q2, k2, v2, mask=mask, causal=causal, striped_ring_attn=striped,
        max_lookback_seq_len=lookback, ring_size=world)

    g = torch.randn_like(out1)
    out1.backward(g)
    out2.backward(g)

    assert (out1 - out2).abs().max().item() < 1e-5
    assert (lse1 - lse2).abs().max().item() < 1e-4
    for a, b_ in ((q1, q2), (k1, k2), (v1, v2)):
        assert (a.grad - b_.grad).abs().max().item() < 1e-5
    return True


@pytest.mark.parametrize("world", [2])
@pytest.mark.parametrize("causal,striped", [(False, False), (True, False), (True, True)])
def test_allgather_vs_ring_w2(world, causal, striped):
    run_distributed(world, _case, causal, striped, 1, False, None)


def test_allgather_vs_ring_w2_gqa_mask():
    run_distributed(2, _case, True, False, 2, True, None)


def test_allgather_vs_ring_w4_striped():
    run_distributed(4, _case, True, True, 1, False, None)


def test_allgather_vs_ring_w2_lookback():
    run_distributed(2, _case, True, False, 1, False, 13)


def test_allgather_vs_ring_w4_gqa_striped_lookback():
    # the full composition at world 4 — closest rehearsal of the 8-GPU
    # bench-adjacent shapes: striped causal + GQA + token-exact lookback
    run_distributed(4, _case, True, True, 2, False, 24)


def _helper_case(rank, world, striped):
    """gather_global_order / scatter_chunks_of_global round-trip on gloo."""
    import torch
    from ring_attention_amd.ops.ring_flash_hip import (
        _gather_global_order, _scatter_chunks_of_global)
    from ring_attention_amd.parallel.collectives import reduce_scatter_chunks
    torch.manual_seed(31)
    b, n, hk, d = 2, 8, 3, 4
    full_ref = torch.arange(b * n * world * hk * d, dtype=torch.float32).reshape(
        b, n * world, hk, d)
    # rank's shard in its layout
    if striped:
        shard = full_ref[:, rank::world]
    else:
        shard = full_ref[:, rank * n:(rank + 1) * n]
    gathered = _gather_global_order(shard.contiguous(), world, striped)
    assert torch.equal(gathered, full_ref), "gather must produce global order"

    # scatter: chunks (W, ...) of a (b,hk,N,d) global tensor; rank's chunk = shard
    g2 = full_ref.permute(0, 2, 1, 3).contiguous()     # (b, hk, N, d)
    chunks = _scatter_chunks_of_global(g2, world, striped, dim=2)
    own_ref = g2[:, :, rank::world] if striped else g2[:, :, rank * n:(rank + 1) * n]
    assert torch.equal(chunks[rank], own_ref)

    # reduce_scatter_chunks (gloo fallback): every rank contributes ones ->
    # own chunk == world * ones
    contrib = torch.ones(world, 4)
    own = reduce_scatter_chunks(contrib)
    assert torch.equal(own, torch.full((4,), float(world)))
    return True


def test_gather_scatter_helpers_contiguous():
    run_distributed(2, _helper_case, False)


def test_gather_scatter_helpers_striped():
    run_distributed(4, _helper_case, True)


def _subring_helper_case(rank, world, ring_size):
    """gather/reduce-scatter within sub-ring process groups (gloo)."""
    import torch
    from ring_attention_amd.parallel import RingTopology
    from ring_attention_amd.parallel.collectives import gather_cat, reduce_scatter_chunks
    topo = RingTopology(ring_size)
    pg = topo.process_group()
    assert (pg is None) == (ring_size == world)

    # each rank contributes a distinct shard; gather within the ring only
    t = torch.full((2, 3), float(rank))
    full = gather_cat(t, dim=0, group=pg)
    expect = torch.cat([torch.full((2, 3), float(topo.ring_base + s))
                        for s in range(ring_size)], dim=0)
    assert torch.equal(full, expect), f"rank {rank}: {full} vs {expect}"

    chunks = torch.ones(ring_size, 4) * (rank + 1)
    own = reduce_scatter_chunks(chunks, group=pg)
    ring_sum = sum(r + 1 for r in range(topo.ring_base, topo.ring_base + ring_size))
    assert torch.equal(own, torch.full((4,), float(ring_sum)))
    return True


def test_subring_collectives_w4r2():
    run_distributed(4, _subring_helper_case, 2)


def test_subring_collectives_w4r4():
    run_distributed(4, _subring_helper_case, 4)
