"""Ring flash attention backed by the CDNA4 HIP kernels (the GPU compute path).

Same semantics as ops/ring_flash.py (the oracle), with an execution-strategy
choice per call (see _choose_strategy):

ALL-GATHER strategy (default when gathered K/V fits memory): one RCCL
all-gather per tensor (stripes across all 7 xGMI links of the full mesh),
ONE local kernel launch over the whole KV (strided-q positions express the
striped layout), and backward ends in a single reduce-scatter for dk/dv.

RING strategy (windows / sub-rings / beyond-memory): double-buffered
overlapped P2P shard circulation; each hop is one resumable kernel launch:
- forward: attn_fwd with (o_acc fp32, m, l) persisting across hops;
  causality, striping and lookback fold into (diag, q_stride, win) per hop;
  fully-masked hops are skipped host-side (kv still circulates).
- backward: delta precomputed once; per hop a row-parallel dq kernel (plain
  fp32 accumulation into one buffer across hops) and a column-parallel dk/dv
  kernel whose per-hop contribution goes through the pipelined
  RingAccumulator in the kernels' transposed scratch layouts
  (dk (B,HK,Nk,D), dv (B,HK,D,Nk)); one final permute at home.

Causal load balance (both strategies): the C++ binding engages paired-tile
scheduling (WG x runs tiles (x, T-1-x) — uniform work) whenever the halved
grid still fills the 256 CUs, combined with FRACTIONAL grid.z splits (each
chunk takes a share of its WG's own valid range) for 256-512-WG grids;
under-filled GQA dkv grids instead use host-built constant-work descriptor
units (_walk_descriptors) with fp32-atomic accumulation.  Env overrides:
RING_ATTN_NO_PAIR, RING_ATTN_NO_DESC, RING_ATTN_KV_SPLIT,
RING_ATTN_SPLIT_DQ/DKV, RING_ATTN_FORCE_STRATEGY, RING_ATTN_AG_BUDGET.

Capability parity with the reference's ring_flash_attn_cuda
(/root/reference/ring_attention_pytorch/ring_flash_attention_cuda.py:40-371)
with its §2.5 bugs fixed.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import Tensor
from torch.autograd import Function

import os

from ..parallel import RingAccumulator, RingTopology, all_ring_pass, is_distributed
from ..parallel.collectives import gather_cat, reduce_scatter_chunks
from . import hip_ext
from .ring_flash import max_hops_for_lookback

# gathered-KV budget for the all-gather strategy (bytes of K+V per rank)
_AG_BUDGET = int(os.environ.get("RING_ATTN_AG_BUDGET", 16 << 30))


def _choose_strategy(use_ring: bool, topo, n, hk, d, lookback, dtype_bytes=2) -> str:
    """'allgather' gathers K/V once over RCCL (stripes across all 7 xGMI
    links; one local kernel; backward ends in ONE reduce-scatter) — the
    MI355X-optimal plan whenever the gathered K/V fits.  'ring' circulates
    shards (used for sliding windows, sub-rings, and beyond-memory seqs)."""
    if not use_ring or topo.ring_size == 1:
        return "local"
    forced = os.environ.get("RING_ATTN_FORCE_STRATEGY")
    if forced in ("ring", "allgather"):
        return forced
    if lookback is not None:
        return "ring"        # window truncates the ring walk — cheaper there
    kv_bytes = 2 * topo.ring_size * n * hk * d * dtype_bytes
    return "allgather" if kv_bytes <= _AG_BUDGET else "ring"


def _gather_global_order(t, ring_size: int, striped: bool, dim: int = 1, group=None):
    """All-gather seq shards and reorder so index == GLOBAL position
    (striped shards interleave: global g = local * R + rank)."""
    full = gather_cat(t, dim=dim, group=group)  # rank-major along dim
    if not striped or ring_size == 1:
        return full
    # rank-major (s, local) -> position-major (local, s)
    shape = t.shape
    n = shape[dim]
    view = full.reshape(*shape[:dim], ring_size, n, *shape[dim + 1:])
    order = list(range(view.ndim))
    order[dim], order[dim + 1] = order[dim + 1], order[dim]
    return view.permute(order).reshape(*shape[:dim], ring_size * n, *shape[dim + 1:]).contiguous()


def _scatter_chunks_of_global(t_full, ring_size: int, striped: bool, dim: int):
    """Inverse mapping for gradients: reshape a global-position-ordered
    tensor into (R, ...) chunks where chunk s is rank s's shard."""
    shape = t_full.shape
    N = shape[dim]
    n = N // ring_size
    if striped:
        view = t_full.reshape(*shape[:dim], n, ring_size, *shape[dim + 1:])
        sdim = dim + 1
    else:
        view = t_full.reshape(*shape[:dim], ring_size, n, *shape[dim + 1:])
        sdim = dim
    order = [sdim] + [i for i in range(view.ndim) if i != sdim]
    return view.permute(order).contiguous()

def _hop_geometry(rq: int, rk: int, n: int, ring_size: int, striped: bool,
                  causal: bool, lookback: int | None) -> tuple[bool, int, int]:
    """Returns (skip_hop, diag, win) for a (q-rank, kv-source-rank) pair,
    in the kernel's qpos semantics: qpos(i) = i * q_stride + diag (q_stride=1
    for ring hops — both sides share the layout stride, which cancels);
    attend(i, j) <=> (not causal or j <= qpos(i)) and (not win or qpos(i) - j <= win).
    """
    if striped:
        diag = 0 if rk <= rq else -1
        if lookback is not None:
            # (i-j)*R + rq-rk <= L  <=>  i-j <= floor((L-rq+rk)/R) = W
            # kernel form: qpos - j = i - j + diag <= W + diag
            win = (lookback - rq + rk) // ring_size + diag
        else:
            win = 0
    else:
        diag = (rq - rk) * n
        win = lookback if lookback is not None else 0   # qpos - j == global distance
    skip = False
    qpos_min, qpos_max = diag, (n - 1) + diag
    if causal and qpos_max < 0:
        skip = True        # whole shard is in the future
    if lookback is not None:
        if qpos_min - (n - 1) > win:
            skip = True    # whole shard beyond the window
        if causal and win < 0:
            skip = True    # window and causal triangle do not intersect
    return skip, diag, win


def _causal_balance_split(causal, lookback, diag_cuts, grid_wgs):
    """Causal load-balance grid.z factor (1 except one measured case).

    Round-1 history: a 2-way grid.z split here gave +19%/+8% at 16k/32k
    causal, but it is SUPERSEDED by in-kernel paired-tile scheduling — the
    C++ binding makes WG x run tiles (x, T-1-x) whenever the paired grid
    still fills the 256 CUs, which measured strictly better wherever either
    engages (c16k 184 vs 228 TF, c32k 231 vs 251) and costs no partial
    buffers or atomics.  RING_ATTN_SPLIT_* env overrides still force
    grid.z splits; RING_ATTN_NO_PAIR disables pairing.

    One case remains python-side: grids in [256, 512) WGs (e.g. the 8k/GPU
    causal shape, 32 qtiles x 8 heads) — pairing alone halves the grid below
    the CU count, so return a 2-way grid.z split; the binding then pairs ON
    TOP of it (paired grid x split = 256+ busy WGs, uniform work, halved
    critical path).
    """
    if causal and lookback is None and diag_cuts and 256 <= grid_wgs < 512:
        return 2
    return 1


_DESC_CACHE = {}


def _walk_descriptors(kind, d, nq, nk, diag, q_stride, bh, device):
    """Constant-work unit descriptors for a causally-cut tile walk.

    Returns int32 (U, 3) [tile, t_lo, t_hi] — or None when the walk is
    empty.  `kind` "dq" = q-tiles (256 rows) walking kv tiles of the dq
    kernel's KVBLK; "dkv" = kv-tiles walking q tiles of QT.  The bounds
    replicate the kernels' causal tile math exactly (no window — desc mode
    is only engaged without lookback).  Uniform unit work removes the
    causal trapezoid imbalance at ANY grid size, with fp32-atomic
    accumulation instead of pairing/splits.
    """
    key = (kind, d, nq, nk, diag, q_stride, bh, str(device))
    hit = _DESC_CACHE.get(key)
    if hit is not None:
        return hit
    if kind == "dq":
        T = (nq + 255) // 256
        W = 128 if d == 64 else 64          # dq_kvblk
        n_w = (nk + W - 1) // W
        bounds = []
        for x in range(T):
            imax = min((x + 1) * 256, nq) - 1
            qmax = imax * q_stride + diag
            hi = 0 if qmax < 0 else min(n_w, qmax // W + 1)
            bounds.append((x, 0, hi))
    else:
        T = (nk + 255) // 256
        W = 64                              # dkv QT (both head dims)
        n_w = (nq + W - 1) // W
        bounds = []
        for x in range(T):
            i_min = -(-(x * 256 - diag) // q_stride)   # ceil div
            lo = 0 if i_min <= 0 else min(n_w, i_min // W)
            bounds.append((x, lo, n_w))
    total = sum(hi - lo for _, lo, hi in bounds)
    if total <= 0:
        desc = None
    else:
        unit = max(1, -(-total // max(32, 768 // max(1, bh))))
        rows = []
        for x, lo, hi in bounds:
            t = lo
            while t < hi:
                rows.append((x, t, min(hi, t + unit)))
                t += unit
        desc = torch.tensor(rows, dtype=torch.int32, device=device)
    _DESC_CACHE[key] = desc
    return desc


def _use_desc(causal, lookback, diag, nk, grid_wgs):
    """Engage descriptor units only where measured to win: a causally-cut
    walk whose natural grid UNDER-FILLS the chip (GQA dkv: kv-tiles x b x hk
    < 256 WGs — GQA causal 16k 185 -> 216 TF with dkv-only descriptors).
    Everywhere else the per-unit atomic output traffic loses to paired-tile
    scheduling (c16k 230 vs 203, causal8k 146 vs 129 when applied to all
    kernels — measured both ways)."""
    return (causal and lookback is None and diag < nk and grid_wgs < 256
            and not os.environ.get("RING_ATTN_NO_DESC"))


# ---------------------------------------------------------------------------
# arbitrary head dims: the gfx950 kernels are instantiated for d in
# {32, 64, 128}; any other d <= 128 is zero-padded to the next instantiated
# size (mathematically exact: zero q/k pad columns contribute nothing to
# QK^T, zero v pad columns are sliced off, and autograd differentiates the
# pad/slice).  Counterpart of the reference's next-pow2 BLOCK_HEADDIM padding
# (triton_flash_attn.py:359,380-421).
# ---------------------------------------------------------------------------
_KERNEL_DIMS = (32, 64, 128)


def kernel_head_dim(d: int) -> int:
    for kd in _KERNEL_DIMS:
        if d <= kd:
            return kd
    raise ValueError(f"head dim {d} > 128 not supported by the gfx950 kernels")


def _pad_head_dim(q, k, v):
    """Zero-pad head dims to the kernel size (the callers pass the TRUE
    d**-0.5 softmax scale through sm_scale, so numerics are exact)."""
    d = q.shape[-1]
    kd = kernel_head_dim(d)
    if kd == d:
        return q, k, v
    return F.pad(q, (0, kd - d)), F.pad(k, (0, kd - d)), F.pad(v, (0, kd - d))


class RingFlashAttentionHIPFunction(Function):
    @staticmethod
    def forward(ctx, q, k, v, mask, causal, bucket_size, ring_reduce, striped,
                max_lookback_seq_len, ring_size, softclamp_qk_sim, softclamp_value,
                sm_scale=None):
        assert q.is_cuda, "HIP path requires GPU tensors"
        b, n, h, d = q.shape
        hk = k.shape[2]
        in_dtype = q.dtype
        scale = sm_scale if sm_scale is not None else d ** -0.5
        lookback = max_lookback_seq_len
        if lookback is not None:
            assert causal, "lookback (sliding window) requires causal"

        use_ring = ring_reduce and is_distributed()
        topo = RingTopology(ring_size if use_ring else 1,
                            rank=None if use_ring else 0,
                            world_size=None if use_ring else 1)
        hops = max_hops_for_lookback(causal, striped, lookback, n, topo.ring_size)

        qb = q.to(torch.bfloat16).contiguous()
        kb = k.to(torch.bfloat16).contiguous()
        vb = v.to(torch.bfloat16).contiguous()
        mask_u8 = mask.to(torch.uint8).contiguous() if mask is not None else None

        ext = hip_ext.require()
        out = torch.empty_like(qb)
        lse = torch.empty(b, h, n, device=q.device, dtype=torch.float32)

        strategy = _choose_strategy(use_ring, topo, n, hk, d, lookback)
        if strategy == "allgather":
            R, rq = topo.ring_size, topo.ring_rank
            pg = topo.process_group()        # sub-ring communicator (or world)
            k_full = _gather_global_order(kb, R, striped, group=pg)
            v_full = _gather_global_order(vb, R, striped, group=pg)
            m_full = _gather_global_order(mask_u8, R, striped, group=pg) if mask_u8 is not None else None
            q_stride = R if striped else 1
            diag = rq if striped else rq * n
            n_total = n * R
            qtiles = (n + 255) // 256
            kv_split = min(16, (n_total + 127) // 128,
                           max(1, 192 // max(1, qtiles * b * h)))
            kv_split = max(kv_split, _causal_balance_split(
                causal, lookback, True, qtiles * b * h))
            kv_split = int(os.environ.get("RING_ATTN_KV_SPLIT", kv_split))
            if kv_split > 1:
                o_part = torch.empty(kv_split, b, h, d, n, device=q.device, dtype=torch.float32)
                m_part = torch.empty(kv_split, b, h, n, device=q.device, dtype=torch.float32)
                l_part = torch.empty(kv_split, b, h, n, device=q.device, dtype=torch.float32)
                ext.attn_fwd(qb, k_full, v_full, m_full, o_part, m_part, l_part,
                             None, None, scale, causal, diag, q_stride,
                             lookback if lookback is not None else 0,
                             lookback is not None,
                             softclamp_qk_sim, softclamp_value, True, True, kv_split, 0, None)
                ext.attn_fwd_merge(o_part, m_part, l_part, None, None, None,
                                   out, lse, kv_split, b, h, d, n, True, True)
            else:
                ext.attn_fwd(qb, k_full, v_full, m_full, None, None, None, out, lse,
                             scale, causal, diag, q_stride,
                             lookback if lookback is not None else 0,
                             lookback is not None,
                             softclamp_qk_sim, softclamp_value, True, True, 1, 0, None)
            ctx.save_for_backward(qb, kb, vb, out, lse,
                                  mask_u8 if mask_u8 is not None else torch.empty(0))
            ctx.sm_scale = scale
            ctx.params = (causal, striped, lookback, hops, softclamp_qk_sim,
                          softclamp_value, use_ring, topo.ring_size, in_dtype,
                          "allgather")
            return out.to(in_dtype), lse

        # kv-split: fill the 256 CUs when the natural grid (q-tiles x b*h) is
        # small (flash-decoding-style partials + a softmax-correct merge)
        qtiles = (n + 255) // 256
        kv_tiles = (n + 63) // 64
        kv_split = min(16, kv_tiles, max(1, 192 // max(1, qtiles * b * h)))
        env_split = os.environ.get("RING_ATTN_KV_SPLIT")

        # which hops actually compute (host-side skip of fully-masked
        # shards); per-hop kv_split: only the hop the causal diagonal cuts
        # through (diag < n) is imbalanced — full hops stay unsplit
        rq = topo.ring_rank
        plan = []
        for hop in range(hops):
            rk = topo.source_of_hop(hop)
            skip, diag, win = _hop_geometry(rq, rk, n, topo.ring_size, striped,
                                            causal, lookback)
            ksp = max(kv_split, _causal_balance_split(
                causal, lookback, diag < n, qtiles * b * h))
            if env_split is not None:
                ksp = int(env_split)
            plan.append((skip, diag, win, ksp))
        max_split = max(p_[3] for p_ in plan)

        multi = hops > 1
        o_acc = m = l = None
        if multi:
            o_acc = torch.empty(b, h, d, n, device=q.device, dtype=torch.float32)
            m = torch.empty(b, h, n, device=q.device, dtype=torch.float32)
            l = torch.empty(b, h, n, device=q.device, dtype=torch.float32)
        o_part = m_part = l_part = None
        if max_split > 1:
            o_part = torch.empty(max_split, b, h, d, n, device=q.device, dtype=torch.float32)
            m_part = torch.empty(max_split, b, h, n, device=q.device, dtype=torch.float32)
            l_part = torch.empty(max_split, b, h, n, device=q.device, dtype=torch.float32)

        active = [i for i, (s, *_) in enumerate(plan) if not s]
        assert active, "every hop masked — degenerate configuration"
        first_active, last_active = active[0], active[-1]

        ring_tensors = (kb, vb) if mask_u8 is None else (kb, vb, mask_u8)

        for info, tensors in all_ring_pass(topo, *ring_tensors, max_hops=hops):
            skip, diag, win, ksp = plan[info.hop]
            if skip:
                continue
            k_t, v_t = tensors[0], tensors[1]
            mk = tensors[2] if mask_u8 is not None else None
            is_f = info.hop == first_active
            is_l = info.hop == last_active
            if ksp > 1:
                ext.attn_fwd(qb, k_t, v_t, mk,
                             o_part, m_part, l_part, None, None,
                             scale, causal, diag, 1, win, lookback is not None,
                             softclamp_qk_sim, softclamp_value,
                             is_f, is_l, ksp, 0, None)
                ext.attn_fwd_merge(o_part, m_part, l_part, o_acc, m, l,
                                   out if is_l else None, lse if is_l else None,
                                   ksp, b, h, d, n, is_f, is_l)
            else:
                ext.attn_fwd(qb, k_t, v_t, mk,
                             o_acc, m, l, out, lse,
                             scale, causal, diag, 1, win, lookback is not None,
                             softclamp_qk_sim, softclamp_value,
                             is_f, is_l, 1, 0, None)

        ctx.save_for_backward(qb, kb, vb, out, lse,
                              mask_u8 if mask_u8 is not None else torch.empty(0))
        ctx.sm_scale = scale
        ctx.params = (causal, striped, lookback, hops, softclamp_qk_sim,
                      softclamp_value, use_ring, topo.ring_size, in_dtype, "ring")
        return out.to(in_dtype), lse

    @staticmethod
    def backward(ctx, do, _dlse):
        qb, kb, vb, out, lse, mask_u8 = ctx.saved_tensors
        mask_u8 = mask_u8 if mask_u8.numel() else None
        (causal, striped, lookback, hops, softclamp_qk_sim, softclamp_value,
         use_ring, ring_size, in_dtype, strategy) = ctx.params
        b, n, h, d = qb.shape
        hk = kb.shape[2]
        scale = ctx.sm_scale

        topo = RingTopology(ring_size if use_ring else 1,
                            rank=None if use_ring else 0,
                            world_size=None if use_ring else 1)
        ext = hip_ext.require()

        dob = do.to(torch.bfloat16).contiguous()
        # delta = rowsum(do * o) in fp32: (b, h, n), fused HIP kernel
        delta = ext.attn_delta(dob, out)

        def _alloc_dq(zeroed):
            fac = torch.zeros if zeroed else torch.empty
            return fac(b, n, h, d, device=qb.device, dtype=torch.float32)

        rq = topo.ring_rank

        if strategy == "allgather":
            R = topo.ring_size
            pg = topo.process_group()
            k_full = _gather_global_order(kb, R, striped, group=pg)
            v_full = _gather_global_order(vb, R, striped, group=pg)
            m_full = _gather_global_order(mask_u8, R, striped, group=pg) if mask_u8 is not None else None
            n_total = n * R
            q_stride = R if striped else 1
            diag = rq if striped else rq * n
            # grid.z splits keep both kernels filling the CUs when their
            # natural grids are small (fp32 atomics, contention = split)
            qtiles = (n + 255) // 256
            kvtiles_t = (n_total + 255) // 256
            split_dq = min(8, max(1, 384 // max(1, qtiles * b * h)))
            split_dkv = min(8, max(1, 384 // max(1, kvtiles_t * b * hk)))
            split_dq = max(split_dq, _causal_balance_split(
                causal, lookback, True, qtiles * b * h))
            split_dkv = max(split_dkv, _causal_balance_split(
                causal, lookback, True, kvtiles_t * b * hk))
            split_dq = int(os.environ.get("RING_ATTN_SPLIT_DQ", split_dq))
            split_dkv = int(os.environ.get("RING_ATTN_SPLIT_DKV", split_dkv))
            ddq = ddkv = None
            if _use_desc(causal, lookback, diag, n_total, kvtiles_t * b * hk):
                ddkv = _walk_descriptors("dkv", d, n, n_total, diag, q_stride,
                                         b * hk, qb.device)
            if ddkv is not None:
                split_dkv = 1
            dq = _alloc_dq(zeroed=split_dq > 1 or ddq is not None)
            fac = torch.zeros if (split_dkv > 1 or ddkv is not None) else torch.empty
            dk_full = fac(b, hk, n_total, d, device=qb.device, dtype=torch.float32)
            dv_full = fac(b, hk, d, n_total, device=qb.device, dtype=torch.float32)
            ext.attn_bwd(qb, k_full, v_full, dob, m_full, lse, delta,
                         dq, dk_full, dv_full, scale, causal, diag, q_stride,
                         lookback if lookback is not None else 0,
                         lookback is not None,
                         softclamp_qk_sim, softclamp_value, False, split_dq, 1,
                         ddq, None)
            ext.attn_bwd(qb, k_full, v_full, dob, m_full, lse, delta,
                         dq, dk_full, dv_full, scale, causal, diag, q_stride,
                         lookback if lookback is not None else 0,
                         lookback is not None,
                         softclamp_qk_sim, softclamp_value, False, split_dkv, 2,
                         None, ddkv)
            # ONE reduce-scatter returns each rank's dk/dv shard (summed)
            dk_chunks = _scatter_chunks_of_global(dk_full, R, striped, dim=2)
            dv_chunks = _scatter_chunks_of_global(dv_full, R, striped, dim=3)
            packed = torch.cat((dk_chunks.reshape(R, -1), dv_chunks.reshape(R, -1)), dim=1)
            own = reduce_scatter_chunks(packed, group=pg)
            half = own.numel() // 2
            dk_home = own[:half].view(b, hk, n, d).permute(0, 2, 1, 3).contiguous()
            dv_home = own[half:].view(b, hk, d, n).permute(0, 3, 1, 2).contiguous()
            return (dq.to(in_dtype), dk_home.to(in_dtype), dv_home.to(in_dtype),
                    None, None, None, None, None, None, None, None, None, None)
        # independent grid.z splits: dq's grid is (q-tiles x b*h), dkv's is
        # (kv-tiles x b*hk) — GQA shrinks the latter (e.g. hk=2 -> 64 WGs)
        qtiles = (n + 255) // 256
        split_dq_base = min(8, max(1, 384 // max(1, qtiles * b * h)))
        split_dkv_base = min(8, max(1, 384 // max(1, qtiles * b * hk)))
        env_dq = os.environ.get("RING_ATTN_SPLIT_DQ")
        env_dkv = os.environ.get("RING_ATTN_SPLIT_DKV")

        # hop plan first: per-hop splits (only the diagonal hop is causally
        # imbalanced; full hops keep the plain fill heuristic) + whether the
        # first compute hop can write dq plainly (no zero-init, no read)
        plan = []
        for hop in range(hops):
            rk_ = topo.source_of_hop(hop)
            skip, diag, win = _hop_geometry(rq, rk_, n, topo.ring_size,
                                            striped, causal, lookback)
            cuts = diag < n
            split_dq = max(split_dq_base, _causal_balance_split(
                causal, lookback, cuts, qtiles * b * h))
            split_dkv = max(split_dkv_base, _causal_balance_split(
                causal, lookback, cuts, qtiles * b * hk))
            if env_dq is not None:
                split_dq = int(env_dq)
            if env_dkv is not None:
                split_dkv = int(env_dkv)
            ddq = ddkv = None
            if not skip and _use_desc(causal, lookback, diag, n,
                                      qtiles * b * hk):
                ddkv = _walk_descriptors("dkv", d, n, n, diag, 1, b * hk,
                                         qb.device)
                if ddkv is not None:
                    split_dkv = 1
            plan.append((skip, diag, win, split_dq, split_dkv, ddq, ddkv))
        first_active = next(i for i, p_ in enumerate(plan) if not p_[0])
        # atomics (grid.z split or descriptor units) need a zeroed base
        dq_zeroed = plan[first_active][3] > 1 or plan[first_active][5] is not None
        dq = _alloc_dq(zeroed=dq_zeroed)

        ring_tensors = (kb, vb) if mask_u8 is None else (kb, vb, mask_u8)
        acc = RingAccumulator(topo)

        for info, tensors in all_ring_pass(topo, *ring_tensors, max_hops=hops):
            skip, diag, win, split_dq, split_dkv, ddq, ddkv = plan[info.hop]
            k_t, v_t = tensors[0], tensors[1]
            mk = tensors[2] if mask_u8 is not None else None
            # circulate dk/dv flat in the kernel's native scratch layouts
            # (dk (b,hk,n,d), dv (b,hk,d,n)); elementwise accumulation is
            # layout-agnostic, so only ONE final permute happens at home.
            # The kernels overwrite every element at split 1, so only the
            # atomic (split/descriptor) and skip (circulated) cases zero-init.
            fac = (torch.zeros if (skip or split_dkv > 1 or ddkv is not None)
                   else torch.empty)
            contrib = fac(2, b * hk * n * d, device=qb.device,
                          dtype=torch.float32)
            if not skip:
                dk_n = contrib[0].view(b, hk, n, d)
                dv_n = contrib[1].view(b, hk, d, n)
                dq_acc_flag = dq_zeroed or info.hop != first_active
                # dq + dk/dv kernels (sequential: their LDS footprints do
                # not co-reside, so stream-splitting buys nothing)
                ext.attn_bwd(qb, k_t, v_t, dob, mk, lse, delta,
                             dq, dk_n, dv_n,
                             scale, causal, diag, 1, win, lookback is not None,
                             softclamp_qk_sim, softclamp_value, dq_acc_flag,
                             split_dq, 1, ddq, None)
                ext.attn_bwd(qb, k_t, v_t, dob, mk, lse, delta,
                             dq, dk_n, dv_n,
                             scale, causal, diag, 1, win, lookback is not None,
                             softclamp_qk_sim, softclamp_value, False,
                             split_dkv, 2, None, ddkv)
            acc.step(contrib, info.is_last)

        dkv = acc.finish(hops)
        dk_home = dkv[0].view(b, hk, n, d).permute(0, 2, 1, 3).contiguous()
        dv_home = dkv[1].view(b, hk, d, n).permute(0, 3, 1, 2).contiguous()

        return (dq.to(in_dtype), dk_home.to(in_dtype), dv_home.to(in_dtype),
                None, None, None, None, None, None, None, None, None, None)


def ring_flash_attn_hip_(
    q: Tensor, k: Tensor, v: Tensor,
    mask: Tensor | None = None,
    causal: bool = False,
    bucket_size: int = 1024,
    ring_reduce_col: bool = False,
    striped_ring_attn: bool = False,
    max_lookback_seq_len: int | None = None,
    ring_size: int | None = None,
    softclamp_qk_sim: bool = False,
    softclamp_value: float = 50.0,
) -> tuple[Tensor, Tensor]:
    d = q.shape[-1]
    sm_scale = None
    if kernel_head_dim(d) != d:
        q, k, v = _pad_head_dim(q, k, v)
        sm_scale = d ** -0.5
    out, lse = RingFlashAttentionHIPFunction.apply(
        q, k, v, mask, causal, bucket_size, ring_reduce_col, striped_ring_attn,
        max_lookback_seq_len, ring_size, softclamp_qk_sim, softclamp_value,
        sm_scale)
    if out.shape[-1] != d:
        out = out[..., :d]
    return out, lse


def ring_flash_attn_hip(q, k, v, **kwargs) -> Tensor:
    out, _ = ring_flash_attn_hip_(q, k, v, **kwargs)
    return out


class FlashAttnOffsetFunction(Function):
    """Local (non-ring) flash attention with a global q-position offset:
    attend(i, j) <=> j <= i + q_offset.  Used by the zig-zag CP scheme, where
    each rank's q chunks are contiguous spans of global positions attending
    the full (all-gathered) KV — one kernel call per span, no score-matrix
    mask materialization (the reference built an explicit O(n*N) bool mask,
    zig_zag_attention.py:123-139)."""

    @staticmethod
    def forward(ctx, q, k, v, q_offset, causal, sm_scale=None):
        assert q.is_cuda
        b, n, h, d = q.shape
        scale = sm_scale if sm_scale is not None else d ** -0.5
        ext = hip_ext.require()
        qb = q.to(torch.bfloat16).contiguous()
        kb = k.to(torch.bfloat16).contiguous()
        vb = v.to(torch.bfloat16).contiguous()
        out = torch.empty_like(qb)
        lse = torch.empty(b, h, n, device=q.device, dtype=torch.float32)
        ext.attn_fwd(qb, kb, vb, None, None, None, None, out, lse,
                     scale, causal, q_offset, 1, 0, False, False, 50.0,
                     True, True, 1, 0, None)
        ctx.save_for_backward(qb, kb, vb, out, lse)
        ctx.meta = (q_offset, causal, q.dtype, scale)
        return out.to(q.dtype)

    @staticmethod
    def backward(ctx, do):
        qb, kb, vb, out, lse = ctx.saved_tensors
        q_offset, causal, in_dtype, scale = ctx.meta
        b, n, h, d = qb.shape
        hk = kb.shape[2]
        nk = kb.shape[1]
        ext = hip_ext.require()
        dob = do.to(torch.bfloat16).contiguous()
        delta = ext.attn_delta(dob, out)
        ddq = ddkv = None
        kvt_ = (nk + 255) // 256
        if _use_desc(causal, None, q_offset, nk, kvt_ * b * hk):
            ddkv = _walk_descriptors("dkv", d, n, nk, q_offset, 1, b * hk, qb.device)
        fac = torch.zeros if ddkv is not None else torch.empty
        dq = torch.empty(b, n, h, d, device=qb.device, dtype=torch.float32)
        dk_n = fac(b, hk, nk, d, device=qb.device, dtype=torch.float32)
        dv_n = fac(b, hk, d, nk, device=qb.device, dtype=torch.float32)
        ext.attn_bwd(qb, kb, vb, dob, None, lse, delta, dq, dk_n, dv_n,
                     scale, causal, q_offset, 1, 0, False, False, 50.0, False, 1, 0,
                     ddq, ddkv)
        dk = dk_n.permute(0, 2, 1, 3).contiguous()
        dv = dv_n.permute(0, 3, 1, 2).contiguous()
        return (dq.to(in_dtype), dk.to(in_dtype), dv.to(in_dtype), None, None, None)


def flash_attn_offset(q, k, v, q_offset=0, causal=True):
    """q (b, n, h, d) attends k/v (b, nk, hk, d) with attend(i,j) <=> j <= i + q_offset."""
    d = q.shape[-1]
    sm_scale = None
    if kernel_head_dim(d) != d:
        q, k, v = _pad_head_dim(q, k, v)
        sm_scale = d ** -0.5
    out = FlashAttnOffsetFunction.apply(q, k, v, q_offset, causal, sm_scale)
    return out[..., :d] if out.shape[-1] != d else out


class FlashAttnFunction(Function):
    """Single-device flash attention with the FULL L0 flag surface of the
    reference kernel pair (triton_flash_attn.py:304-430, 988-1128): additive
    bias (vector (b,h,nk) or matrix (b,h,n,nk)), causal with optional strict
    diagonal (striped attention), key-pad mask, tanh softclamp and sliding
    window.  bias gradients are not produced (parity: the reference backward
    has no db either)."""

    @staticmethod
    def forward(ctx, q, k, v, bias, bias_is_matrix, key_pad_mask, causal,
                causal_mask_diagonal, softclamp, softclamp_value, window,
                sm_scale=None):
        assert q.is_cuda
        b, n, h, d = q.shape
        scale = sm_scale if sm_scale is not None else d ** -0.5
        ext = hip_ext.require()
        qb = q.to(torch.bfloat16).contiguous()
        kb = k.to(torch.bfloat16).contiguous()
        vb = v.to(torch.bfloat16).contiguous()
        biasf = bias.float().contiguous() if bias is not None else None
        mk = key_pad_mask.to(torch.uint8).contiguous() if key_pad_mask is not None else None
        out = torch.empty_like(qb)
        lse = torch.empty(b, h, n, device=q.device, dtype=torch.float32)
        diag = -1 if (causal and causal_mask_diagonal) else 0
        ext.attn_fwd(qb, kb, vb, mk, None, None, None, out, lse,
                     scale, causal, diag, 1,
                     window if window is not None else 0, window is not None,
                     softclamp, softclamp_value, True, True, 1, 0, None,
                     bias=biasf, bias_mat=bool(bias_is_matrix))
        ctx.save_for_backward(qb, kb, vb, out, lse,
                              *( (biasf,) if biasf is not None else () ),
                              *( (mk,) if mk is not None else () ))
        ctx.meta = (bias is not None, bias_is_matrix, key_pad_mask is not None,
                    causal, diag, softclamp, softclamp_value, window, q.dtype,
                    scale)
        return out.to(q.dtype)

    @staticmethod
    def backward(ctx, do):
        (has_bias, bias_mat, has_mask, causal, diag, softclamp,
         softclamp_value, window, in_dtype, scale) = ctx.meta
        saved = list(ctx.saved_tensors)
        qb, kb, vb, out, lse = saved[:5]
        rest = saved[5:]
        biasf = rest.pop(0) if has_bias else None
        mk = rest.pop(0) if has_mask else None
        b, n, h, d = qb.shape
        hk = kb.shape[2]
        nk = kb.shape[1]
        ext = hip_ext.require()
        dob = do.to(torch.bfloat16).contiguous()
        delta = ext.attn_delta(dob, out)
        dq = torch.empty(b, n, h, d, device=qb.device, dtype=torch.float32)
        dk_n = torch.empty(b, hk, nk, d, device=qb.device, dtype=torch.float32)
        dv_n = torch.empty(b, hk, d, nk, device=qb.device, dtype=torch.float32)
        ext.attn_bwd(qb, kb, vb, dob, mk, lse, delta, dq, dk_n, dv_n,
                     scale, causal, diag, 1,
                     window if window is not None else 0, window is not None,
                     softclamp, softclamp_value, False, 1, 0, None, None,
                     bias=biasf, bias_mat=bool(bias_mat))
        dk = dk_n.permute(0, 2, 1, 3).contiguous()
        dv = dv_n.permute(0, 3, 1, 2).contiguous()
        return (dq.to(in_dtype), dk.to(in_dtype), dv.to(in_dtype),
                None, None, None, None, None, None, None, None, None)


def flash_attn(
    q: Tensor, k: Tensor, v: Tensor,
    bias: Tensor | None = None,
    key_pad_mask: Tensor | None = None,
    causal: bool = False,
    causal_mask_diagonal: bool = False,
    softclamp_qk_sim: bool = False,
    softclamp_value: float = 50.0,
    window: int | None = None,
) -> Tensor:
    """Single-device flash attention, (b, n, h, d) layout, any d <= 128.

    ``bias``: additive attention bias in the natural-log domain, shaped
    (b, h, nk) (broadcast over query rows) or (b, h, n, nk).  GQA: q heads h
    pair kv heads via qh % hk (reference tile convention)."""
    d = q.shape[-1]
    bias_is_matrix = bias is not None and bias.dim() == 4
    if bias is not None:
        bh = q.shape[2]
        if bias_is_matrix:
            bias = bias.expand(q.shape[0], bh, q.shape[1], k.shape[1])
        else:
            bias = bias.expand(q.shape[0], bh, k.shape[1])
    sm_scale = None
    if kernel_head_dim(d) != d:
        q, k, v = _pad_head_dim(q, k, v)
        sm_scale = d ** -0.5
    out = FlashAttnFunction.apply(q, k, v, bias, bias_is_matrix, key_pad_mask,
                                  causal, causal_mask_diagonal,
                                  softclamp_qk_sim, softclamp_value, window,
                                  sm_scale)
    return out[..., :d] if out.shape[-1] != d else out
