"""MX-FP8 serving forward — an MI355X-only capability beyond the reference.

The reference's kernels are bf16/fp16 Triton
(/root/reference/ring_attention_pytorch/triton_flash_attn.py); gfx950 adds
block-scaled e4m3 MFMA at ~2x the bf16 matrix rate, which this module
exposes as a quantize + fused-forward pair for inference/prefill:

    out, lse = flash_attn_fp8(q, k, v)          # (b, n, h, d) bf16 in/out

Numerics: Q/K use one e8m0 scale per row (2^ceil(log2(amax/448))), V one
per (d row, 64-kv chunk), and the softmax matrix P needs no scale at all
(exp2(x - m) <= 1).  Softmax and the output accumulator stay fp32 inside
the kernel; only the MFMA operands are 8-bit.  Measured error vs an fp32
reference: ~5% mean-relative on out (P's e4m3 quantization alone accounts
for ~3.4% — tools/fp8_dbg.py), lse within 0.03 on long rows; use the bf16
path when training.

Causal (``causal=True``) uses the bf16 kernels' mirrored paired-tile
load balance; GQA follows the framework-wide ``qh % hk`` pairing.

Scope: forward only (use the bf16 path for training); no bias/key-pad
mask/window; any head dim <= 128 (native 64/128 + exact zero-pad).
Ragged lengths are handled by padding the quantized buffers (zero bytes —
a valid e4m3, never NaN) while the kernel masks at the true kv length.
"""

from __future__ import annotations

import torch
from torch import Tensor

from . import hip_ext


def _e8m0(amax: Tensor) -> tuple[Tensor, Tensor]:
    """amax (positive fp32) -> (e8m0 byte = exp+127, scale = 2^exp)."""
    e = torch.ceil(torch.log2(amax.clamp(min=2.0 ** -126) / 448.0))
    e = e.clamp(-127.0, 127.0)
    return (e + 127.0).to(torch.uint8), torch.exp2(e)


def _pad_dim(x: Tensor, dim: int, mult: int) -> Tensor:
    n = x.shape[dim]
    pad = (-n) % mult
    if pad == 0:
        return x.contiguous()
    shape = list(x.shape)
    shape[dim] = pad
    return torch.cat((x, x.new_zeros(shape)), dim=dim).contiguous()


def quantize_fp8(q: Tensor, k: Tensor, v: Tensor):
    """Quantize (b, n, h, d) q/k/v for attn_fwd_fp8.

    Returns (q8, k8, v8t, qs, ks, vs) uint8 views, PADDED to the kernel's
    tile alignment (q rows to 256, kv rows to 128) with zero bytes and
    scale 2^-127 — zero is a valid e4m3 (never NaN), and the kernel masks
    scores at the true kv length (nk_true):
      q8/k8 (b, nq_pad, h, d) e4m3; v8t (b, hk, d, nk_pad) e4m3;
      qs/ks (b, n_pad, h, d // 64) e8m0; vs (b, hk, d, nk_pad // 64) e8m0.
    """
    qf, kf = q.float(), k.float()
    d = q.shape[-1]
    assert d % 64 == 0
    nd = d // 64
    # per (row, 64-d chunk) scales: shape (..., nd)
    qs_b, qs_s = _e8m0(qf.view(*qf.shape[:-1], nd, 64).abs().amax(dim=-1))
    ks_b, ks_s = _e8m0(kf.view(*kf.shape[:-1], nd, 64).abs().amax(dim=-1))
    q8 = (qf / qs_s.repeat_interleave(64, dim=-1)).to(torch.float8_e4m3fn)
    k8 = (kf / ks_s.repeat_interleave(64, dim=-1)).to(torch.float8_e4m3fn)

    vt = v.permute(0, 2, 3, 1).float().contiguous()      # (b, hk, d, n)
    b, hk, dv, n = vt.shape
    n64 = n + ((-n) % 64)
    vt = _pad_dim(vt, 3, 128)
    vs_b, vs_s = _e8m0(vt[..., :n64].view(b, hk, dv, n64 // 64, 64)
                       .abs().amax(dim=-1))
    vs_b = _pad_dim(vs_b, 3, 2)          # to nk_pad // 64
    vs_s = _pad_dim(vs_s, 3, 2).clamp(min=2.0 ** -127)
    v8t = (vt / vs_s.repeat_interleave(64, dim=-1)).to(torch.float8_e4m3fn)
    return (_pad_dim(q8.view(torch.uint8), 1, 256),
            _pad_dim(k8.view(torch.uint8), 1, 128),
            v8t.view(torch.uint8).contiguous(),
            _pad_dim(qs_b, 1, 256), _pad_dim(ks_b, 1, 128),
            vs_b.contiguous())


@torch.no_grad()
def flash_attn_fp8(
    q: Tensor, k: Tensor, v: Tensor,
    sm_scale: float | None = None,
    causal: bool = False,
) -> tuple[Tensor, Tensor]:
    """MX-FP8 attention forward on (b, n, h, d) tensors.

    Returns (out bf16 (b, n, h, d), lse fp32 (b, h, n)).  Quantizes
    internally; pass pre-quantized operands via flash_attn_fp8_quantized
    to amortize quantization across decode steps.
    """
    d = q.shape[-1]
    sm = sm_scale if sm_scale is not None else d ** -0.5
    if d % 64 != 0:
        # exact zero-pad to the next kernel dim ({64, 128}, like the bf16
        # path's _pad_head_dim): padded q/k dims contribute 0 to scores,
        # padded v dims are sliced off below
        kd = 64 if d < 64 else 128
        assert d <= 128, "fp8 path: head dim <= 128"
        import torch.nn.functional as F
        q = F.pad(q, (0, kd - d))
        k = F.pad(k, (0, kd - d))
        v = F.pad(v, (0, kd - d))
    q8, k8, v8t, qs, ks, vs = quantize_fp8(q, k, v)
    out, lse = flash_attn_fp8_quantized(
        q8, k8, v8t, qs, ks, vs, sm,
        causal=causal, nk_true=k.shape[1])
    nq = q.shape[1]
    return out[:, :nq, :, :d], lse[..., :nq]


def flash_attn_fp8_quantized(q8, k8, v8t, qs, ks, vs, sm_scale: float,
                             causal: bool = False, nk_true: int = 0):
    if q8.is_cuda:
        out, lse = hip_ext.require().attn_fwd_fp8(q8, k8, v8t, qs, ks, vs,
                                                  sm_scale, causal, nk_true)
        return out, lse
    return _eager_fp8(q8, k8, v8t, qs, ks, vs, sm_scale, causal, nk_true)


def _dequant(x8: Tensor, scales: Tensor) -> Tensor:
    return (x8.view(torch.float8_e4m3fn).float()
            * torch.exp2(scales.float() - 127.0).repeat_interleave(64, dim=-1))


def _eager_fp8(q8, k8, v8t, qs, ks, vs, sm_scale, causal, nk_true):
    """CPU fallback: dequantized eager attention with the kernel's masking
    semantics (true-length + causal).  Approximates the GPU path to within
    its P-quantization error (the fallback keeps P in fp32) — it exists so
    the fp8 wrapper/ring logic runs and tests anywhere, like the rest of
    the framework's CPU fallbacks."""
    b, nq, h, d = q8.shape
    hk, nk = k8.shape[2], k8.shape[1]
    if nk_true <= 0:
        nk_true = nk
    qd = _dequant(q8, qs)                                  # (b, nq, h, d)
    kd = _dequant(k8, ks)                                  # (b, nk, hk, d)
    # v8t (b, hk, d, nk), vs (b, hk, d, nk//64): dequant along the last dim
    vd = (v8t.view(torch.float8_e4m3fn).float()
          * torch.exp2(vs.float() - 127.0).repeat_interleave(64, dim=-1))
    vd = vd.permute(0, 3, 1, 2)                            # (b, nk, hk, d)
    groups = h // hk
    if groups > 1:
        kd = kd.repeat(1, 1, groups, 1)                    # qh pairs qh % hk
        vd = vd.repeat(1, 1, groups, 1)
    sim = torch.einsum("bihd,bjhd->bhij", qd, kd) * sm_scale
    jpos = torch.arange(nk, device=sim.device)
    sim = sim.masked_fill((jpos >= nk_true)[None, None, None, :], float("-inf"))
    if causal:
        ipos = torch.arange(nq, device=sim.device)
        sim = sim.masked_fill((jpos[None, :] > ipos[:, None])[None, None],
                              float("-inf"))
    lse = sim.logsumexp(dim=-1)                            # (b, h, nq)
    out = torch.einsum("bhij,bjhd->bihd", torch.softmax(sim, dim=-1), vd)
    out = torch.nan_to_num(out)                            # all-masked rows
    return out.to(torch.bfloat16), lse.float()


@torch.no_grad()
def ring_flash_attn_fp8(
    q: Tensor, k: Tensor, v: Tensor,
    sm_scale: float | None = None,
    ring_size: int | None = None,
    causal: bool = False,
) -> tuple[Tensor, Tensor]:
    """RING attention forward in MX-FP8 over sharded KV.

    Each rank quantizes its LOCAL (b, n_shard, hk, d) k/v once; the 8-bit
    shards then rotate the ring — half the bf16 wire bytes per hop (xGMI
    ring hops are single-link bound, so the fp8 wire format directly halves
    the transport time too) — and every hop's fused-fp8 partial merges by
    logsumexp (exact online-softmax combine in fp32).

    Causal (rank-ordered shards, rank r owning global rows
    [r*n_shard, (r+1)*n_shard)): a hop whose source ring-rank is ahead of
    this rank contributes nothing and skips the kernel entirely; the
    diagonal hop (source == self) runs the causal fp8 kernel; earlier
    sources run the full non-causal kernel.  Striped/zig-zag layouts stay
    on the bf16 ring path.

    Returns (out bf16 (b, n, h, d), lse fp32 (b, h, n)).
    """
    from ..parallel import RingTopology, all_ring_pass, is_distributed
    from ..parallel.ring_pass import null_ring_pass

    sm = sm_scale if sm_scale is not None else q.shape[-1] ** -0.5
    q8, k8, v8t, qs, ks, vs = quantize_fp8(q, k, v)

    my_rank = 0
    if is_distributed():
        topo = RingTopology(ring_size)
        my_rank = topo.ring_rank
        hops = all_ring_pass(topo, k8, v8t, ks, vs)
    else:
        hops = null_ring_pass(k8, v8t, ks, vs)

    out = lse = None
    for info, (k8h, v8th, ksh, vsh) in hops:
        src = info.source_ring_rank
        if causal and src > my_rank:
            continue                                  # strictly future rows
        o_h, l_h = flash_attn_fp8_quantized(
            q8, k8h, v8th, qs, ksh, vsh, sm,
            causal=causal and src == my_rank, nk_true=k.shape[1])
        if out is None:
            out, lse = o_h.float(), l_h
        else:
            m = torch.maximum(lse, l_h)
            wa = (lse - m).exp()                      # (b, h, n)
            wb = (l_h - m).exp()
            den = wa + wb
            wa4 = (wa / den).permute(0, 2, 1).unsqueeze(-1)   # (b, n, h, 1)
            wb4 = (wb / den).permute(0, 2, 1).unsqueeze(-1)
            out = out * wa4 + o_h.float() * wb4
            lse = m + den.log()
    nq = q.shape[1]
    return out[:, :nq].to(q.dtype), lse[..., :nq]


def quantize_kv_cache(k: Tensor, v: Tensor):
    """Quantize a decode KV cache (b, hk, n, d) to e4m3 + per-row e8m0.

    Returns (k8, v8, ks, vs) uint8: k8/v8 (b, hk, n, d), ks/vs (b, hk, n).
    Halves the decode kernel's HBM stream (decode is bandwidth-bound); the
    scales fold into the score / softmax weight, so the decode math is the
    same fp32 online softmax.  Quantize once at cache-write time.
    """
    kf, vf = k.float(), v.float()
    ks_b, ks_s = _e8m0(kf.abs().amax(dim=-1))
    vs_b, vs_s = _e8m0(vf.abs().amax(dim=-1))
    k8 = (kf / ks_s.unsqueeze(-1)).to(torch.float8_e4m3fn)
    v8 = (vf / vs_s.unsqueeze(-1)).to(torch.float8_e4m3fn)
    return (k8.view(torch.uint8), v8.view(torch.uint8),
            ks_b.contiguous(), vs_b.contiguous())
