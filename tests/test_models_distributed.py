"""RingAttention / RingTransformer distributed ≡ replicated non-ring twin.

The reference proved the same equivalences with its assert.py / assert_attn.py
CLI scripts; here they are pytest-able, run on gloo, and check
gradients through the full model with an explicit all-reduce average (the
DDP-equivalent) against the replicated ground truth.
"""

import pytest
import torch
import torch.distributed as dist

from ring_attention_amd import RingAttention, RingTransformer

from .distributed_utils import run_distributed


def _attn_case(rank, world, causal, striped, groups, batch_size, seq_len, rotary):
    torch.manual_seed(7)
    dim, heads, dim_head = 32, 4, 16
    bucket_size = 8
    ring_seq_size = 16  # per-rank shard size; chunks = padded_seq / 16 must divide world

    kwargs = dict(
        dim=dim, dim_head=dim_head, heads=heads,
        num_grouped_query_heads=groups, causal=causal,
        bucket_size=bucket_size, ring_seq_size=ring_seq_size,
        striped_ring_attn=striped, rotary_embed=rotary,
        use_hip_kernel=False,
    )
    ring_model = RingAttention(ring_attn=True, auto_shard_seq=True, **kwargs)
    flat_model = RingAttention(ring_attn=False, auto_shard_seq=False, **kwargs)
    flat_model.load_state_dict(ring_model.state_dict())

    torch.manual_seed(100)  # same full input on every rank
    full_x = torch.randn(batch_size * world, seq_len, dim)
    xs = full_x.chunk(world, dim=0)[rank].clone().requires_grad_(True)
    x_flat = full_x.clone().requires_grad_(True)

    out = ring_model(xs)
    ref = flat_model(x_flat)
    ref_shard = ref.chunk(world, dim=0)[rank]

    out_err = (out - ref_shard).abs().max().item()
    assert out_err < 2e-5, f"out err {out_err}"

    g = torch.randn_like(ref)
    out.backward(g.chunk(world, dim=0)[rank])
    ref.backward(g)

    din_err = (xs.grad - x_flat.grad.chunk(world, dim=0)[rank]).abs().max().item()
    assert din_err < 2e-5, f"input grad err {din_err}"

    # parameter grads: DDP-average across ranks == full-batch grad / world... the
    # ring loss is per-rank on its own batch shard, so sum of rank grads == full grad
    for (name, p_ring), (_, p_flat) in zip(ring_model.named_parameters(),
                                           flat_model.named_parameters()):
        g_sum = p_ring.grad.clone()
        dist.all_reduce(g_sum)
        perr = (g_sum - p_flat.grad).abs().max().item()
        assert perr < 5e-5, f"param grad err {name}: {perr}"
    return out_err


def test_attn_world2_causal():
    run_distributed(2, _attn_case, True, False, 1, 1, 32, False)


def test_attn_world2_striped_rotary():
    run_distributed(2, _attn_case, True, True, 1, 1, 32, True)


def test_attn_world2_gqa_padded():
    # seq 27 -> padded to 32, exercises mask synthesis
    run_distributed(2, _attn_case, True, False, 2, 1, 27, False)


def test_attn_world4_subrings():
    # seq 32 -> 2 chunks of 16; world 4 => 2 sharded batches (2 independent rings)
    run_distributed(4, _attn_case, True, False, 1, 1, 32, False)


def test_attn_world2_noncausal():
    run_distributed(2, _attn_case, False, False, 1, 1, 32, False)


def _transformer_case(rank, world, causal, striped, groups, seq_len, lookback):
    torch.manual_seed(11)
    model_kwargs = dict(
        num_tokens=64, dim=32, depth=2, causal=causal, dim_head=16, heads=4,
        ff_mult=2, num_grouped_query_heads=groups, bucket_size=8,
        ring_seq_size=16, striped_ring_attn=striped,
        max_lookback_seq_len=lookback, use_hip_kernel=False,
    )
    ring_model = RingTransformer(ring_attn=True, **model_kwargs)
    flat_model = RingTransformer(ring_attn=False, **model_kwargs)
    flat_model.load_state_dict(ring_model.state_dict())

    torch.manual_seed(200)
    full_ids = torch.randint(0, 64, (world, seq_len))
    ids = full_ids[rank:rank + 1]

    # inference path: logits gathered back to batch shards
    logits = ring_model(ids)
    ref_logits = flat_model(full_ids)
    lerr = (logits - ref_logits[rank:rank + 1]).abs().max().item()
    assert lerr < 5e-5, f"logits err {lerr}"

    # loss path + embedding grads (DDP-style average == replicated grad / ...)
    loss = ring_model(ids, return_loss=True)
    loss.backward()
    ref_loss = flat_model(full_ids, return_loss=True)
    ref_loss.backward()

    g_sum = ring_model.token_emb.weight.grad.clone()
    dist.all_reduce(g_sum)
    g_sum /= world
    gerr = (g_sum - flat_model.token_emb.weight.grad).abs().max().item()
    assert gerr < 5e-5, f"emb grad err {gerr}"

    loss_avg = loss.detach().clone()
    dist.all_reduce(loss_avg)
    loss_avg /= world
    assert abs(loss_avg.item() - ref_loss.item()) < 1e-4
    return lerr


def test_transformer_world2_causal():
    run_distributed(2, _transformer_case, True, False, 1, 33, None)


def test_transformer_world2_striped_gqa():
    run_distributed(2, _transformer_case, True, True, 2, 33, None)


def test_transformer_world2_lookback():
    run_distributed(2, _transformer_case, True, False, 1, 33, 16)


def test_transformer_world4():
    run_distributed(4, _transformer_case, True, False, 1, 65, None)


def _var_batch_case(rank, world):
    """Ranks contribute different batch sizes (reference's --batch-size-var-len)."""
    torch.manual_seed(13)
    model_kwargs = dict(
        num_tokens=64, dim=32, depth=1, causal=True, dim_head=16, heads=2,
        bucket_size=8, ring_seq_size=16, use_hip_kernel=False,
    )
    ring_model = RingTransformer(ring_attn=True, **model_kwargs)
    flat_model = RingTransformer(ring_attn=False, **model_kwargs)
    flat_model.load_state_dict(ring_model.state_dict())

    torch.manual_seed(300)
    sizes = [1 + (r % 2) for r in range(world)]           # e.g. [1, 2]
    full_ids = torch.randint(0, 64, (sum(sizes), 32))
    start = sum(sizes[:rank])
    ids = full_ids[start:start + sizes[rank]]

    logits = ring_model(ids)
    ref_logits = flat_model(full_ids)
    err = (logits - ref_logits[start:start + sizes[rank]]).abs().max().item()
    assert err < 5e-5, f"var-batch logits err {err}"
    return err


def test_transformer_var_batch_world2():
    run_distributed(2, _var_batch_case)


def _presharded_case(rank, world, striped):
    """Statically pre-sharded data (utils/data.py) == the model's auto-shard path."""
    from ring_attention_amd.utils.data import shard_sequence_batch
    torch.manual_seed(17)
    kwargs = dict(num_tokens=64, dim=32, depth=1, causal=True, dim_head=16,
                  heads=2, bucket_size=8, ring_seq_size=16,
                  striped_ring_attn=striped, use_hip_kernel=False)
    auto_model = RingTransformer(ring_attn=True, auto_shard_seq=True, **kwargs)
    pre_model = RingTransformer(ring_attn=True, auto_shard_seq=False, **kwargs)
    pre_model.load_state_dict(auto_model.state_dict())

    torch.manual_seed(400)
    ids = torch.randint(0, 64, (2, 29))           # padded to 32 -> 2 shards of 16

    sb = shard_sequence_batch(ids, ring_seq_size=16, bucket_size=8,
                              world=world, rank=rank, striped=striped)
    loss_pre = pre_model(sb.ids, mask=sb.mask, labels=sb.labels, return_loss=True)
    assert torch.isfinite(loss_pre)

    # ground truth: replicated non-ring model over the full (unsharded) batch;
    # its shard-restricted CE must equal the pre-sharded ring loss
    flat_model = RingTransformer(ring_attn=False, **kwargs)
    flat_model.load_state_dict(auto_model.state_dict())
    x_full, lab_full = ids[:, :-1], ids[:, 1:]
    logits_full = flat_model(x_full)              # (b, 28, vocab)
    import torch.nn.functional as F
    from ring_attention_amd.utils.sharding import stripe_permute as sp
    n = x_full.shape[1]
    pad = 32 - n
    logits_pad = torch.nn.functional.pad(logits_full, (0, 0, 0, pad))
    labels_pad = torch.nn.functional.pad(lab_full, (0, pad), value=-1)
    if striped:
        logits_pad = sp(logits_pad, world)
        labels_pad = sp(labels_pad, world)
    sl = slice(rank * 16, (rank + 1) * 16)
    ref_loss = F.cross_entropy(logits_pad[:, sl].permute(0, 2, 1),
                               labels_pad[:, sl], ignore_index=-1)
    assert abs(loss_pre.item() - ref_loss.item()) < 1e-4, (
        f"pre-sharded loss {loss_pre.item()} vs ref {ref_loss.item()}")
    return float(loss_pre.detach())


def test_presharded_data_world2():
    run_distributed(2, _presharded_case, False)


def test_presharded_data_world2_striped():
    run_distributed(2, _presharded_case, True)
