import os, sys, torch
sys.path.insert(0, "/root/repo"); sys.path.insert(0, "/root/repo/tests")
from loopback_dist import loopback_world
from ring_attention_amd.ops.ring_flash_hip import ring_flash_attn_hip_

def case(world, name, n_total=8192, h=8, causal=True, **env):
    b, d = 1, 64
    for k_, v_ in env.items(): os.environ[k_] = v_
    os.environ["RING_ATTN_FORCE_STRATEGY"] = "ring"
    try:
        torch.manual_seed(17)
        q = torch.randn(b, n_total, h, d, device="cuda", dtype=torch.bfloat16)
        k = torch.randn_like(q); v = torch.randn_like(q)
        g = torch.randn_like(q)
        qr = q.clone().requires_grad_(True)
        kr = k.clone().requires_grad_(True)
        vr = v.clone().requires_grad_(True)
        ref, _ = ring_flash_attn_hip_(qr, kr, vr, causal=causal)
        ref.backward(g)
        n = n_total // world
        def run(rank):
            sl = slice(rank * n, (rank + 1) * n)
            qs = q[:, sl].clone().requires_grad_(True)
            ks = k[:, sl].clone().requires_grad_(True)
            vs = v[:, sl].clone().requires_grad_(True)
            out, _ = ring_flash_attn_hip_(qs, ks, vs, causal=causal,
                                          ring_reduce_col=True, ring_size=world)
            out.backward(g[:, sl])
            return qs.grad, ks.grad, vs.grad
        res = loopback_world(world, run)
        msg = []
        for rank, (dq, dk, dv) in enumerate(res):
            sl = slice(rank * n, (rank + 1) * n)
            es = []
            for got, want in ((dq, qr.grad[:, sl]), (dk, kr.grad[:, sl]), (dv, vr.grad[:, sl])):
                es.append((got.float() - want.float()).abs().max().item()
                          / (want.float().abs().max().item() + 1e-6))
            msg.append(f"r{rank}:" + ",".join(f"{e:.2g}" for e in es))
        print(name, " ".join(msg))
    finally:
        for k_ in list(env) + ["RING_ATTN_FORCE_STRATEGY"]: os.environ.pop(k_, None)

case(4, "w4 default")
case(4, "w4 split1", RING_ATTN_SPLIT_DQ="1", RING_ATTN_SPLIT_DKV="1")
case(4, "w4 split1+nodesc", RING_ATTN_SPLIT_DQ="1", RING_ATTN_SPLIT_DKV="1", RING_ATTN_NO_DESC="1")
case(4, "w4 nodesc", RING_ATTN_NO_DESC="1")
case(4, "w4 nopair", RING_ATTN_NO_PAIR="1")
case(4, "w4 noncausal", causal=False)
case(2, "w2 default")
