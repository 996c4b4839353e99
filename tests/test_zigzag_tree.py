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
