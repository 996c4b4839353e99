"""Eager O(n^2) ground-truth attention — the test oracle.

Capability parity with the reference's ``default_attention``
(/root/reference/ring_attention_pytorch/ring_attention.py:47-98): GQA, causal,
key-padding mask, tanh softclamp.  This is the semantic definition every other
path (bucketed ring flash, HIP kernels) is tested against.

Layout: q (b, n, h, d); k, v (b, n_kv, h_kv, d) with h % h_kv == 0.
"""

from __future__ import annotations


import torch
from torch import Tensor

MASK_VALUE = torch.finfo(torch.float32).min


def softclamp(t: Tensor, value: float) -> Tensor:
    return (t / value).tanh() * value


def default_attention(
    q: Tensor,
    k: Tensor,
    v: Tensor,
    mask: Tensor | None = None,        # (b, n_kv) bool; True = attend
    causal: bool = False,
    softclamp_qk_sim: bool = False,
    softclamp_value: float = 50.0,
    q_positions: Tensor | None = None,  # (n,) global positions of q rows (for permuted layouts)
    k_positions: Tensor | None = None,  # (n_kv,) global positions of k rows
    max_lookback_seq_len: int | None = None,  # sliding window (token-exact)
) -> Tensor:
    """Exact attention in fp32.  ``*_positions`` generalize the causal mask to
    permuted (striped / zig-zag) sequence layouts: row i may attend col j iff
    k_positions[j] <= q_positions[i] (default: identity positions); the
    optional window masks q_pos - k_pos > max_lookback_seq_len."""
    b, n, h, d = q.shape
    _, nk, hk, _ = k.shape
    assert h % hk == 0
    groups = h // hk
    out_dtype = q.dtype

    q, k, v = q.float(), k.float(), v.float()
    if groups > 1:
        # reference GQA convention ('... h d -> ... (g h) d', ring_attention.py:86-89):
        # q head qh pairs kv head qh % hk (tile, not repeat_interleave)
        k = k.repeat(1, 1, groups, 1)
        v = v.repeat(1, 1, groups, 1)

    scale = d ** -0.5
    sim = torch.einsum("bihd,bjhd->bhij", q, k) * scale

    if softclamp_qk_sim:
        sim = softclamp(sim, softclamp_value)

    if mask is not None:
        sim = sim.masked_fill(~mask[:, None, None, :], MASK_VALUE)

    if causal or max_lookback_seq_len is not None:
        qp = q_positions if q_positions is not None else torch.arange(n, device=q.device)
        kp = k_positions if k_positions is not None else torch.arange(nk, device=q.device)
        if causal:
            causal_mask = kp[None, :] > qp[:, None]      # (n, nk): True = masked
            sim = sim.masked_fill(causal_mask[None, None, :, :], MASK_VALUE)
        if max_lookback_seq_len is not None:
            window_mask = (qp[:, None] - kp[None, :]) > max_lookback_seq_len
            sim = sim.masked_fill(window_mask[None, None, :, :], MASK_VALUE)

    attn = sim.softmax(dim=-1)
    out = torch.einsum("bhij,bjhd->bihd", attn, v)
    return out.to(out_dtype)
