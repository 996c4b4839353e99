import os, sys, torch
sys.path.insert(0, "/root/repo")
os.environ["RING_ATTN_FWD_V2"] = "1"
from ring_attention_amd.ops import hip_ext
ext = hip_ext.require()
b, n, h, d = 1, 128, 1, 128
torch.manual_seed(0)
q = torch.randn(b, n, h, d, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
out = torch.empty_like(q)
lse = torch.empty(b, h, n, device="cuda", dtype=torch.float32)
ext.attn_fwd(q, k, v, None, None, None, None, out, lse, d ** -0.5,
             False, 0, 1, 0, False, False, 50.0, True, True, 1, 0, None)
torch.cuda.synchronize()
qf, kf, vf = q.float(), k.float(), v.float()
sim = torch.einsum("bihd,bjhd->bhij", qf, kf) * d ** -0.5
ref = torch.einsum("bhij,bjhd->bihd", sim.softmax(-1), vf)
err = (out.float() - ref).abs()[0, :, 0, :]    # (n, d)
print("max err", err.max().item())
rows = (err.max(dim=1).values > 0.05).nonzero().flatten().tolist()
cols = (err.max(dim=0).values > 0.05).nonzero().flatten().tolist()
print("bad rows:", rows[:40], "..." if len(rows) > 40 else "", f"({len(rows)})")
print("bad cols:", cols[:40], "..." if len(cols) > 40 else "", f"({len(cols)})")
lse_ref = sim.logsumexp(-1)[0, 0]
print("lse err", (lse[0, 0] - lse_ref).abs().max().item())
