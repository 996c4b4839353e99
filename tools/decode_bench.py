"""Tree-decode step timing (BASELINE config 5 shape, single GPU portion).
128k KV sharded over N GPUs => at N=1 measure the full-seq local partial."""
import sys, time, torch
sys.path.insert(0, "/root/repo")
from ring_attention_amd.tree_decode import tree_attn_decode
import sys
b, h, d = 1, 8, 64
n = int(sys.argv[1]) if len(sys.argv) > 1 else 131072
torch.manual_seed(0)
q = torch.randn(b, h, 1, d, device="cuda", dtype=torch.bfloat16)
k = torch.randn(b, h, n, d, device="cuda", dtype=torch.bfloat16)
v = torch.randn(b, h, n, d, device="cuda", dtype=torch.bfloat16)
for _ in range(5):
    out = tree_attn_decode(q, k, v, shard_kv_seq=False)
torch.cuda.synchronize(); t0 = time.perf_counter()
iters = 50
for _ in range(iters):
    out = tree_attn_decode(q, k, v, shard_kv_seq=False)
torch.cuda.synchronize()
us = (time.perf_counter() - t0) / iters * 1e6
kv_gb = 2 * b * h * n * d * 2 / 1e9
print(f"decode step: {us:.1f} us for {n} KV ({kv_gb:.2f} GB -> {kv_gb/us*1e6:.0f} GB/s)")
# correctness spot check
sim = torch.einsum("bhid,bhjd->bhij", q.float(), k.float()) * d ** -0.5
ref = torch.einsum("bhij,bhjd->bhid", sim.softmax(-1), v.float())
print("err:", (out.float() - ref).abs().max().item())
