import os, sys, torch
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/tests")
from loopback_dist import loopback_world
from ring_attention_amd.ops.ring_flash_hip import ring_flash_attn_hip_

def shard_idx(n_total, world, rank, striped):
    if striped:
        return torch.arange(n_total // world, device="cuda") * world + rank
    return torch.arange(n_total // world, device="cuda") + rank * (n_total // world)

def case(world, striped, hk, lookback, strategy, n_total=8192, h=8):
    b, d = 1, 64
    torch.manual_seed(17)
    q = torch.randn(b, n_total, h, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, n_total, hk, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, n_total, hk, d, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(b, n_total, h, d, device="cuda", dtype=torch.bfloat16)
    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ref, _ = ring_flash_attn_hip_(qr, kr, vr, causal=True, max_lookback_seq_len=lookback)
    ref.backward(g)
    os.environ["RING_ATTN_FORCE_STRATEGY"] = strategy
    try:
        def run(rank):
            idx = shard_idx(n_total, world, rank, striped)
            qs = q[:, idx].clone().requires_grad_(True)
            ks = k[:, idx].clone().requires_grad_(True)
            vs = v[:, idx].clone().requires_grad_(True)
            out, _ = ring_flash_attn_hip_(qs, ks, vs, causal=True,
                                          ring_reduce_col=True, striped_ring_attn=striped,
                                          max_lookback_seq_len=lookback, ring_size=world)
            out.backward(g[:, idx])
            return out.detach(), qs.grad, ks.grad, vs.grad
        res = loopback_world(world, run)
    finally:
        del os.environ["RING_ATTN_FORCE_STRATEGY"]
    worst = {}
    for rank, (out, dq, dk, dv) in enumerate(res):
        idx = shard_idx(n_total, world, rank, striped)
        for got, want, name in ((out, ref.detach()[:, idx], "out"), (dq, qr.grad[:, idx], "dq"),
                                (dk, kr.grad[:, idx], "dk"), (dv, vr.grad[:, idx], "dv")):
            e = (got.float() - want.float()).abs().max().item() / (want.float().abs().max().item() + 1e-6)
            worst[name] = max(worst.get(name, 0), e)
    print(f"w{world} striped={striped} hk={hk} lb={lookback} {strategy}: " +
          " ".join(f"{k}={v:.3g}" for k, v in worst.items()))

for args in [(8, True, 8, None, "ring"), (8, True, 8, None, "allgather"),
             (8, False, 8, None, "ring"), (8, True, 2, None, "ring"),
             (8, True, 8, 2048, "ring"), (4, True, 8, 2048, "ring"),
             (2, True, 8, 2048, "ring"), (8, True, 2, 2048, "allgather")]:
    try:
        case(*args)
    except Exception as ex:
        print(args, "EXC:", str(ex)[:120])
