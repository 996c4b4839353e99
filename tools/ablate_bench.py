"""Diagnostic: forward kernel with staging ablated (stage-once) vs full.
Usage (GPU box): python tools/ablate_bench.py"""
import sys, time, torch
sys.path.insert(0, "/root/repo")
from ring_attention_amd.ops import hip_ext
ext = hip_ext.require()
b, n, h, d = 1, 8192, 8, 64
torch.manual_seed(0)
q = torch.randn(b, n, h, d, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
out = torch.empty_like(q)
lse = torch.empty(b, h, n, device="cuda", dtype=torch.float32)
scale = d ** -0.5
def run(ablate, iters=30):
    for _ in range(5):
        ext.attn_fwd(q, k, v, None, None, None, None, out, lse, scale,
                     False, 0, 1, 0, False, False, 50.0, True, True, 1, ablate, None)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters):
        ext.attn_fwd(q, k, v, None, None, None, None, out, lse, scale,
                     False, 0, 1, 0, False, False, 50.0, True, True, 1, ablate, None)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6
print(f"full kernel:  {run(0):8.1f} us")
print(f"stage-once:   {run(1):8.1f} us")
print(f"no-softmax:   {run(2):8.1f} us")
