"""End-to-end distributed training example for ring_attention_amd.

One process per GPU over RCCL:

    torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \\
        examples/train_ring_transformer.py --seq-len 65536 --steps 50

Shows the full framework surface: RingTransformer with striped ring
attention + GQA + rotary, timeout-guarded init, the progress watchdog,
DDP gradient sync, tracing ranges, and checkpoint/resume.
Synthetic data (random tokens) — swap `make_batch` for a real loader.
"""

from __future__ import annotations

import argparse
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from ring_attention_amd import RingTransformer
from ring_attention_amd.parallel.watchdog import Watchdog, init_distributed
from ring_attention_amd.utils.checkpoint import load_checkpoint, save_checkpoint
from ring_attention_amd.utils.tracing import trace_range


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--seq-len", type=int, default=8192)
    ap.add_argument("--dim", type=int, default=512)
    ap.add_argument("--depth", type=int, default=4)
    ap.add_argument("--heads", type=int, default=8)
    ap.add_argument("--gqa-groups", type=int, default=2)
    ap.add_argument("--vocab", type=int, default=50000)
    ap.add_argument("--lr", type=float, default=3e-4)
    ap.add_argument("--ckpt", type=str, default="")
    ap.add_argument("--save-every", type=int, default=0)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        torch.cuda.set_device(local_rank)
    if world > 1:
        init_distributed(timeout_s=300)

    device = torch.device("cuda", local_rank) if on_gpu else torch.device("cpu")
    torch.manual_seed(7)

    shard = max(512, args.seq_len // max(world, 1))
    model = RingTransformer(
        num_tokens=args.vocab, dim=args.dim, depth=args.depth, causal=True,
        dim_head=64, heads=args.heads, num_grouped_query_heads=args.gqa_groups,
        bucket_size=min(shard, 1024), ring_seq_size=shard,
        ring_attn=world > 1, striped_ring_attn=world > 1,
    ).to(device)
    if on_gpu:
        model = model.bfloat16()

    opt = torch.optim.AdamW(model.parameters(), lr=args.lr)
    start_step = 0
    if args.ckpt and os.path.exists(args.ckpt):
        meta = load_checkpoint(args.ckpt, model, opt)
        start_step = meta["step"]
        if rank == 0:
            print(f"resumed from {args.ckpt} at step {start_step}")

    def make_batch():
        return torch.randint(0, args.vocab, (1, args.seq_len + 1), device=device)

    # DDP for gradient averaging; the ring attention composes with it (the
    # reference demonstrated the same composition, assert.py:97-98)
    net = model
    if world > 1:
        net = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank] if on_gpu else None)

    wd = Watchdog(stall_s=120).start()
    t0 = time.perf_counter()
    for step in range(start_step, args.steps):
        with trace_range(f"train_step_{step}"):
            ids = make_batch()
            loss = net(ids, return_loss=True)
            loss.backward()
            opt.step()
            opt.zero_grad(set_to_none=True)
        wd.tick(step)
        if rank == 0 and step % 10 == 0:
            el = time.perf_counter() - t0
            tok_s = (step - start_step + 1) * args.seq_len * max(world, 1) / el
            print(f"step {step}  loss {loss.item():.4f}  {tok_s:,.0f} tok/s")
        if args.save_every and args.ckpt and (step + 1) % args.save_every == 0:
            save_checkpoint(args.ckpt, model, opt, step=step + 1)
    wd.stop()

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
