"""Zig-zag CP and tree-attention decode distributed equivalence tests.

Mirrors the reference's assert_zig_zag.py / assert_tree_attn.py coverage
 with per-element comparisons against a replicated ground truth.
"""

import torch

from ring_attention_amd.ops.reference import MASK_VALUE
from ring_attention_amd.tree_decode import tree_attn_decode
from ring_attention_amd.zigzag import zig_zag_attn, zig_zag_pad_seq, zig_zag_shard

from .distributed_utils import run_distributed


def _zigzag_case(rank, world, seq_len, groups):
    torch.manual_seed(5)
    b, h, d = 2, 4, 16
    hk = h // groups
    q_full = torch.randn(b, h, seq_len, d)
    k_full = torch.randn(b, hk, seq_len, d)
    v_full = torch.randn(b, hk, seq_len, d)

    q_pad, inv_pad = zig_zag_pad_seq(q_full)
    k_pad, _ = zig_zag_pad_seq(k_full)
    v_pad, _ = zig_zag_pad_seq(v_full)
    n_pad = q_pad.shape[-2]

    (q_loc, q_idx, kv_idx), inverse = zig_zag_shard(q_pad)
    (k_loc, _, _), _ = zig_zag_shard(k_pad)
    (v_loc, _, _), _ = zig_zag_shard(v_pad)

    # causal mask from exported positions (pad keys masked out)
    valid = kv_idx < seq_len
    attn_mask = (kv_idx[None, :] <= q_idx[:, None]) & valid[None, :]
    q_loc = q_loc.requires_grad_(True)
    out = zig_zag_attn(q_loc, k_loc, v_loc, attn_mask=attn_mask[None, None])

    out_full = inverse(out)
    out_full = inv_pad(out_full)

    # replicated ground truth: plain causal attention (b h n d layout)
    kk = k_full.repeat(1, groups, 1, 1)
    vv = v_full.repeat(1, groups, 1, 1)
    sim = torch.einsum("bhid,bhjd->bhij", q_full, kk) * d ** -0.5
    pos = torch.arange(seq_len)
    sim = sim.masked_fill((pos[None, :] > pos[:, None])[None, None], MASK_VALUE)
    ref = torch.einsum("bhij,bhjd->bhid", sim.softmax(-1), vv)

    err = (out_full - ref).abs().max().item()
    assert err < 1e-5, f"zigzag out err {err}"
    return err


def test_zigzag_world2():
    run_distributed(2, _zigzag_case, 37, 1)


def test_zigzag_world2_gqa():
    run_distributed(2, _zigzag_case, 64, 2)


def test_zigzag_world4():
    run_distributed(4, _zigzag_case, 57, 1)


def _tree_case(rank, world, seq_len):
    torch.manual_seed(3)
    b, h, d = 2, 4, 32
    q = torch.randn(b, h, 1, d)
    k = torch.randn(b, h, seq_len, d)
    v = torch.randn(b, h, seq_len, d)

    out = tree_attn_decode(q, k, v, shard_kv_seq=True)

    sim = torch.einsum("bhid,bhjd->bhij", q, k) * d ** -0.5
    ref = torch.einsum("bhij,bhjd->bhid", sim.softmax(-1), v)
    err = (out - ref).abs().max().item()
    assert err < 1e-5, f"tree decode err {err}"
    return err


def test_tree_decode_world2():
    run_distributed(2, _tree_case, 37)


def test_tree_decode_world4():
    run_distributed(4, _tree_case, 64)


def test_tree_decode_world4_short_seq():
    # seq shorter than world: some ranks hold no KV (edge case)
    run_distributed(4, _tree_case, 3)


def _zigzag_causal_derived_case(rank, world, seq_len, groups):
    # portable path with causal=True and NO attn_mask: the mask must be
    # derived from q_chunk_starts (silently non-causal before)
    torch.manual_seed(11)
    b, h, d = 1, 2 * groups, 16
    hk = h // groups
    q_full = torch.randn(b, h, seq_len, d)
    k_full = torch.randn(b, hk, seq_len, d)
    v_full = torch.randn(b, hk, seq_len, d)

    q_pad, inv_pad = zig_zag_pad_seq(q_full)
    k_pad, _ = zig_zag_pad_seq(k_full)
    v_pad, _ = zig_zag_pad_seq(v_full)

    (q_loc, q_idx, _), inverse = zig_zag_shard(q_pad)
    (k_loc, _, _), _ = zig_zag_shard(k_pad)
    (v_loc, _, _), _ = zig_zag_shard(v_pad)

    half = q_loc.shape[-2] // 2
    starts = (int(q_idx[0]), int(q_idx[half]))
    out = zig_zag_attn(q_loc, k_loc, v_loc, causal=True,
                       q_chunk_starts=starts, kv_valid_len=seq_len)
    out_full = inv_pad(inverse(out))

    kk = k_full.repeat(1, groups, 1, 1)
    vv = v_full.repeat(1, groups, 1, 1)
    sim = torch.einsum("bhid,bhjd->bhij", q_full, kk) * d ** -0.5
    pos = torch.arange(seq_len)
    sim = sim.masked_fill((pos[None, :] > pos[:, None])[None, None], MASK_VALUE)
    ref = torch.einsum("bhij,bhjd->bhid", sim.softmax(-1), vv)
    err = (out_full - ref).abs().max().item()
    assert err < 1e-5, f"zigzag causal-derived err {err}"
    return err


def test_zigzag_causal_derived_world2():
    run_distributed(2, _zigzag_causal_derived_case, 37, 1)


def test_zigzag_causal_derived_world2_gqa():
    run_distributed(2, _zigzag_causal_derived_case, 64, 2)


def test_zigzag_causal_requires_positions():
    import pytest
    q = torch.randn(1, 2, 8, 16)
    with pytest.raises(ValueError):
        zig_zag_attn(q, q, q, causal=True)


def _cross_strategy_gqa_case(rank, world, seq_len, groups):
    # ring and zig-zag must pair GQA heads IDENTICALLY (the
    # reference tile convention, qh -> qh % hk) so the same weights give the
    # same outputs whichever CP strategy runs.
    from ring_attention_amd.ops.reference import default_attention
    from ring_attention_amd.ops.ring_flash import ring_flash_attn

    torch.manual_seed(13)
    b, h, d = 1, 4, 16
    hk = h // groups
    q_full = torch.randn(b, seq_len, h, d)       # b n h d layout (ring)
    k_full = torch.randn(b, seq_len, hk, d)
    v_full = torch.randn(b, seq_len, hk, d)

    # oracle (tile convention by definition)
    ref = default_attention(q_full, k_full, v_full, causal=True)

    # ring path on the sharded sequence
    sl = slice(rank * seq_len // world, (rank + 1) * seq_len // world)
    out_ring = ring_flash_attn(q_full[:, sl], k_full[:, sl], v_full[:, sl],
                               causal=True, ring_reduce_col=True,
                               bucket_size=seq_len // (2 * world))
    err_ring = (out_ring - ref[:, sl]).abs().max().item()
    assert err_ring < 1e-5, f"ring GQA err {err_ring}"

    # zig-zag path on the same tensors ((b h n d) layout)
    qz, kz, vz = (t.permute(0, 2, 1, 3) for t in (q_full, k_full, v_full))
    (q_loc, q_idx, kv_idx), inverse = zig_zag_shard(qz)
    (k_loc, _, _), _ = zig_zag_shard(kz)
    (v_loc, _, _), _ = zig_zag_shard(vz)
    attn_mask = kv_idx[None, :] <= q_idx[:, None]
    out_z = zig_zag_attn(q_loc, k_loc, v_loc, attn_mask=attn_mask[None, None])
    out_zf = inverse(out_z).permute(0, 2, 1, 3)
    err_z = (out_zf - ref).abs().max().item()
    assert err_z < 1e-5, f"zigzag GQA err {err_z}"
    return max(err_ring, err_z)


def test_cross_strategy_gqa_world2():
    run_distributed(2, _cross_strategy_gqa_case, 32, 2)


def _tree_case_dv(rank, world, seq_len, dv):
    # value dim != head dim, including ranks with empty shards
    torch.manual_seed(7)
    b, h, d = 1, 2, 16
    q = torch.randn(b, h, 1, d)
    k = torch.randn(b, h, seq_len, d)
    v = torch.randn(b, h, seq_len, dv)
    out = tree_attn_decode(q, k, v, shard_kv_seq=True)
    sim = torch.einsum("bhid,bhjd->bhij", q, k) * d ** -0.5
    ref = torch.einsum("bhij,bhjd->bhid", sim.softmax(-1), v)
    err = (out - ref).abs().max().item()
    assert err < 1e-5, f"tree decode dv err {err}"
    return err


def test_tree_decode_dv_mismatch_world4_short_seq():
    # empty-shard ranks must size the packed buffer from v's dv
    run_distributed(4, _tree_case_dv, 3, 24)


def _tree_case_multiq_gqa(rank, world, seq_len, nq, groups):
    # generalized decode: nq query tokens per head + GQA kv heads
    torch.manual_seed(23)
    b, h, d = 1, 4, 16
    hk = h // groups
    q = torch.randn(b, h, nq, d)
    k = torch.randn(b, hk, seq_len, d)
    v = torch.randn(b, hk, seq_len, d)
    out = tree_attn_decode(q, k, v, shard_kv_seq=True)
    kk = k.repeat(1, groups, 1, 1)
    vv = v.repeat(1, groups, 1, 1)
    sim = torch.einsum("bhid,bhjd->bhij", q, kk) * d ** -0.5
    ref = torch.einsum("bhij,bhjd->bhid", sim.softmax(-1), vv)
    err = (out - ref).abs().max().item()
    assert err < 1e-5, f"tree decode multiq/gqa err {err}"
    return err


def test_tree_decode_multiquery_world2():
    run_distributed(2, _tree_case_multiq_gqa, 64, 4, 1)


def test_tree_decode_gqa_world2():
    run_distributed(2, _tree_case_multiq_gqa, 64, 2, 2)
