"""Multi-rank composition tests via the in-process loopback dist layer.

CPU tests run the fp32 oracle ring path under `loopback_world` and compare
against the unsharded local oracle — this validates the loopback harness
itself (same collectives call-pattern as the GPU path).

GPU tests (-m gpu) run the REAL HIP multi-rank composition — ring P2P +
RingAccumulator homecoming, all-gather-KV strategy with striped gather
order, sub-ring process groups, packed dk/dv reduce-scatter, tree-decode
all-reduces — on ONE MI355X, which real RCCL refuses ("Duplicate GPU
detected").  This is the only pre-driver coverage of the world>1 HIP path.
"""

import os

import pytest
import torch

from .loopback_dist import loopback_world


def _shard_idx(n_total, world, rank, striped, device="cpu"):
    if striped:
        return torch.arange(n_total // world, device=device) * world + rank
    return torch.arange(n_total // world, device=device) + rank * (n_total // world)


# --------------------------------------------------------------------------
# CPU: oracle ring under loopback vs local oracle (validates the harness)
# --------------------------------------------------------------------------

@pytest.mark.parametrize("world,striped,hk", [(2, False, 4), (2, True, 4),
                                              (4, False, 2), (4, True, 4)])
def test_loopback_oracle_ring_cpu(world, striped, hk):
    from ring_attention_amd.ops.ring_flash import ring_flash_attn_
    b, n_total, h, d = 2, 256, 4, 32
    n = n_total // world
    torch.manual_seed(3)
    q = torch.randn(b, n_total, h, d)
    k = torch.randn(b, n_total, hk, d)
    v = torch.randn(b, n_total, hk, d)
    g = torch.randn(b, n_total, h, d)

    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ref, _ = ring_flash_attn_(qr, kr, vr, causal=True, bucket_size=64)
    ref.backward(g)

    def run(rank):
        idx = _shard_idx(n_total, world, rank, striped)
        qs = q[:, idx].clone().requires_grad_(True)
        ks = k[:, idx].clone().requires_grad_(True)
        vs = v[:, idx].clone().requires_grad_(True)
        out, _ = ring_flash_attn_(qs, ks, vs, causal=True, bucket_size=64,
                                  ring_reduce_col=True,
                                  striped_ring_attn=striped, ring_size=world)
        out.backward(g[:, idx])
        return out.detach(), qs.grad, ks.grad, vs.grad

    results = loopback_world(world, run)
    for rank, (out, dq, dk, dv) in enumerate(results):
        idx = _shard_idx(n_total, world, rank, striped)
        for got, want, name in ((out, ref.detach()[:, idx], "out"),
                                (dq, qr.grad[:, idx], "dq"),
                                (dk, kr.grad[:, idx], "dk"),
                                (dv, vr.grad[:, idx], "dv")):
            e = (got - want).abs().max().item()
            assert e < 1e-4, f"rank {rank} {name} err {e}"


def test_loopback_subring_oracle_cpu():
    """world 4 split into two rings of 2: each ring attends only its own
    half-sequence; compare against per-ring local oracles."""
    from ring_attention_amd.ops.ring_flash import ring_flash_attn_
    world, ring_size = 4, 2
    b, n, h, d = 1, 64, 2, 16
    torch.manual_seed(5)
    shards = torch.randn(world, b, n, h, d)

    def run(rank):
        qs = shards[rank].clone().requires_grad_(True)
        out, _ = ring_flash_attn_(qs, qs.detach(), qs.detach(), causal=True,
                                  bucket_size=32, ring_reduce_col=True,
                                  ring_size=ring_size)
        return out.detach()

    results = loopback_world(world, run)
    for ring in range(world // ring_size):
        members = range(ring * ring_size, (ring + 1) * ring_size)
        full = torch.cat([shards[r] for r in members], dim=1)
        ref, _ = ring_flash_attn_(full, full, full, causal=True, bucket_size=32)
        for i, r in enumerate(members):
            want = ref[:, i * n:(i + 1) * n]
            e = (results[r] - want).abs().max().item()
            assert e < 1e-4, f"ring {ring} rank {r} err {e}"


# --------------------------------------------------------------------------
# GPU: the real HIP multi-rank composition on one MI355X
# --------------------------------------------------------------------------

@pytest.mark.gpu
@pytest.mark.parametrize("strategy,causal,striped,hk,lookback", [
    ("allgather", False, False, 4, None),
    ("allgather", True, False, 4, None),
    ("allgather", True, True, 4, None),
    ("allgather", True, False, 2, None),     # GQA kv circulation
    ("ring", True, False, 4, None),
    ("ring", False, False, 4, None),
    ("ring", True, True, 4, None),
    ("ring", True, False, 4, 1024),          # lookback truncates the walk
])
def test_loopback_hip_multirank(strategy, causal, striped, hk, lookback):
    from ring_attention_amd.ops.ring_flash_hip import ring_flash_attn_hip_
    world = 2
    b, n_total, h, d = 2, 4096, 4, 64
    torch.manual_seed(11)
    q = torch.randn(b, n_total, h, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, n_total, hk, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, n_total, hk, d, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(b, n_total, h, d, device="cuda", dtype=torch.bfloat16)

    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ref, _ = ring_flash_attn_hip_(qr, kr, vr, causal=causal,
                                  max_lookback_seq_len=lookback)
    ref.backward(g)

    os.environ["RING_ATTN_FORCE_STRATEGY"] = strategy
    try:
        def run(rank):
            idx = _shard_idx(n_total, world, rank, striped, device="cuda")
            qs = q[:, idx].clone().requires_grad_(True)
            ks = k[:, idx].clone().requires_grad_(True)
            vs = v[:, idx].clone().requires_grad_(True)
            out, _ = ring_flash_attn_hip_(qs, ks, vs, causal=causal,
                                          ring_reduce_col=True,
                                          striped_ring_attn=striped,
                                          max_lookback_seq_len=lookback,
                                          ring_size=world)
            out.backward(g[:, idx])
            return out.detach(), qs.grad, ks.grad, vs.grad

        results = loopback_world(world, run)
    finally:
        del os.environ["RING_ATTN_FORCE_STRATEGY"]

    for rank, (out, dq, dk, dv) in enumerate(results):
        idx = _shard_idx(n_total, world, rank, striped, device="cuda")
        for got, want, name in ((out, ref.detach()[:, idx], "out"),
                                (dq, qr.grad[:, idx], "dq"),
                                (dk, kr.grad[:, idx], "dk"),
                                (dv, vr.grad[:, idx], "dv")):
            e = (got.float() - want.float()).abs().max().item()
            s = want.float().abs().max().item() + 1e-6
            assert e / s < 4e-2, f"rank {rank} {name} rel err {e/s}"


@pytest.mark.gpu
def test_loopback_hip_subring_allgather():
    """world 4, ring_size 2 — sub-ring process groups drive the gather and
    the dk/dv reduce-scatter; each ring must match its own local HIP run."""
    from ring_attention_amd.ops.ring_flash_hip import ring_flash_attn_hip_
    world, ring_size = 4, 2
    b, n, h, d = 1, 1024, 4, 64
    torch.manual_seed(13)
    shards = torch.randn(world, b, n, h, d, device="cuda", dtype=torch.bfloat16)
    grads = torch.randn(world, b, n, h, d, device="cuda", dtype=torch.bfloat16)

    refs = []
    for ring in range(world // ring_size):
        members = list(range(ring * ring_size, (ring + 1) * ring_size))
        full = torch.cat([shards[r] for r in members], dim=1).requires_grad_(True)
        out, _ = ring_flash_attn_hip_(full, full.detach(), full.detach(),
                                      causal=True)
        out.backward(torch.cat([grads[r] for r in members], dim=1))
        refs.append((out.detach(), full.grad))

    os.environ["RING_ATTN_FORCE_STRATEGY"] = "allgather"
    try:
        def run(rank):
            qs = shards[rank].clone().requires_grad_(True)
            out, _ = ring_flash_attn_hip_(qs, qs.detach(), qs.detach(),
                                          causal=True, ring_reduce_col=True,
                                          ring_size=ring_size)
            out.backward(grads[rank])
            return out.detach(), qs.grad

        results = loopback_world(world, run)
    finally:
        del os.environ["RING_ATTN_FORCE_STRATEGY"]

    for rank, (out, dq) in enumerate(results):
        ring, pos = rank // ring_size, rank % ring_size
        ref_out, _ = refs[ring]
        want = ref_out[:, pos * n:(pos + 1) * n]
        e = (out.float() - want.float()).abs().max().item()
        s = want.float().abs().max().item() + 1e-6
        assert e / s < 4e-2, f"rank {rank} out rel err {e/s}"


@pytest.mark.gpu
def test_loopback_tree_decode():
    from ring_attention_amd.tree_decode import tree_attn_decode
    world = 2
    b, h, d, n = 2, 8, 64, 8192
    torch.manual_seed(17)
    q = torch.randn(b, h, 1, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, h, n, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, h, n, d, device="cuda", dtype=torch.bfloat16)
    ref = tree_attn_decode(q, k, v, shard_kv_seq=False)

    def run(rank):
        ks = k.chunk(world, dim=-2)[rank]
        vs = v.chunk(world, dim=-2)[rank]
        return tree_attn_decode(q, ks, vs, shard_kv_seq=False)

    results = loopback_world(world, run)
    for rank, out in enumerate(results):
        e = (out.float() - ref.float()).abs().max().item()
        s = ref.float().abs().max().item() + 1e-6
        assert e / s < 2e-2, f"rank {rank} rel err {e/s}"
