"""Distributed equivalence checker — the reference's assert.py / assert_attn.py
UX (spawn a real multi-process world, compare against a replicated ground
truth) for this framework.  CPU/gloo by default so it runs anywhere; pass
--use-gpu on a multi-GPU box for RCCL.

    python examples/assert_equivalence.py --world-size 4 --causal
    python examples/assert_equivalence.py --world-size 2 --striped --groups 2
    python examples/assert_equivalence.py --world-size 4 --model   # RingTransformer

(Reference counterpart: /root/reference assert.py:141-197, assert_attn.py.)
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.multiprocessing as mp


def _setup(rank, world, use_gpu):
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29521")
    backend = "nccl" if use_gpu else "gloo"
    if use_gpu:
        torch.cuda.set_device(rank)
    torch.distributed.init_process_group(backend, rank=rank, world_size=world)


def _attn_case(rank, world, args):
    _setup(rank, world, args.use_gpu)
    from ring_attention_amd.ops.reference import default_attention
    from ring_attention_amd.ops.ring_flash import ring_flash_attn_

    device = torch.device("cuda", rank) if args.use_gpu else "cpu"
    b, h, d = args.batch, args.heads, args.dim_head
    hk = h // args.groups
    n_total = args.seq_len - args.seq_len % (world * args.bucket_size) \
        or world * args.bucket_size
    n = n_total // world
    torch.manual_seed(42)
    q = torch.randn(b, n_total, h, d, device=device)
    k = torch.randn(b, n_total, hk, d, device=device)
    v = torch.randn(b, n_total, hk, d, device=device)
    g = torch.randn(b, n_total, h, d, device=device)

    if args.striped:
        idx = torch.arange(n, device=device) * world + rank
    else:
        idx = torch.arange(n, device=device) + rank * n

    qs = q[:, idx].clone().requires_grad_(True)
    ks = k[:, idx].clone().requires_grad_(True)
    vs = v[:, idx].clone().requires_grad_(True)
    out, _ = ring_flash_attn_(qs, ks, vs, causal=args.causal,
                              bucket_size=args.bucket_size,
                              ring_reduce_col=True,
                              striped_ring_attn=args.striped,
                              ring_size=world)
    out.backward(g[:, idx])

    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ref = default_attention(qr, kr, vr, causal=args.causal)
    ref.backward(g)

    ok = True
    for got, want, name in ((out, ref.detach()[:, idx], "out"),
                            (qs.grad, qr.grad[:, idx], "dq"),
                            (ks.grad, kr.grad[:, idx], "dk"),
                            (vs.grad, vr.grad[:, idx], "dv")):
        e = (got - want).abs().max().item()
        s = want.abs().max().item() + 1e-9
        good = e / s < (2e-2 if args.use_gpu else 1e-5)
        ok &= good
        if rank == 0:
            print(f"  {name}: rel err {e / s:.2e}  {'OK' if good else 'FAIL'}")
    if rank == 0:
        print("PASS" if ok else "FAIL", flush=True)
    torch.distributed.destroy_process_group()
    assert ok


def _model_case(rank, world, args):
    _setup(rank, world, args.use_gpu)
    from ring_attention_amd import RingTransformer

    device = torch.device("cuda", rank) if args.use_gpu else "cpu"
    torch.manual_seed(7)
    kwargs = dict(num_tokens=256, dim=64, depth=2, causal=args.causal,
                  dim_head=args.dim_head, heads=args.heads,
                  bucket_size=args.bucket_size)
    ring = RingTransformer(ring_attn=True, striped_ring_attn=args.striped,
                           ring_seq_size=world * args.bucket_size, **kwargs).to(device)
    flat = RingTransformer(ring_attn=False, **kwargs).to(device)
    flat.load_state_dict(ring.state_dict())

    seq = torch.randint(0, 256, (2, 31), device=device)   # awkward length
    logits_ring = ring(seq)
    logits_flat = flat(seq)
    e = (logits_ring - logits_flat).abs().max().item()
    good = e < (2e-2 if args.use_gpu else 1e-4)
    if rank == 0:
        print(f"  RingTransformer logits max err {e:.2e}  "
              f"{'PASS' if good else 'FAIL'}", flush=True)
    torch.distributed.destroy_process_group()
    assert good


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--world-size", type=int, default=2)
    ap.add_argument("--seq-len", type=int, default=512)
    ap.add_argument("--bucket-size", type=int, default=32)
    ap.add_argument("--batch", type=int, default=2)
    ap.add_argument("--heads", type=int, default=4)
    ap.add_argument("--dim-head", type=int, default=32)
    ap.add_argument("--groups", type=int, default=1, help="GQA q-head groups")
    ap.add_argument("--causal", action="store_true")
    ap.add_argument("--striped", action="store_true")
    ap.add_argument("--model", action="store_true",
                    help="compare a RingTransformer against a non-ring twin")
    ap.add_argument("--use-gpu", action="store_true")
    args = ap.parse_args()
    fn = _model_case if args.model else _attn_case
    mp.spawn(fn, args=(args.world_size, args), nprocs=args.world_size, join=True)


if __name__ == "__main__":
    main()
