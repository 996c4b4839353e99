"""Multi-query decode latency: the VERDICT r1 #9 acceptance measurement
(>= 2 queries/step at < 1.3x single-query latency)."""
import sys, time, torch
sys.path.insert(0, "/root/repo")
from ring_attention_amd.tree_decode import tree_attn_decode
b, h, n, d = 1, 8, 131072, 64
torch.manual_seed(0)
k = torch.randn(b, h, n, d, device="cuda", dtype=torch.bfloat16)
v = torch.randn_like(k)
for nq in (1, 2, 4, 8):
    q = torch.randn(b, h, nq, d, device="cuda", dtype=torch.bfloat16)
    for _ in range(5):
        tree_attn_decode(q, k, v, shard_kv_seq=False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
        tree_attn_decode(q, k, v, shard_kv_seq=False)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / 50 * 1e6
    tb = b * h * n * d * 2 * 2 / (us * 1e-6) / 1e12
    print(f"nq={nq}: {us:8.1f} us/step  kv-stream {tb:5.2f} TB/s")
# GQA: 32q/4kv heads
k4 = torch.randn(b, 4, n, d, device="cuda", dtype=torch.bfloat16)
v4 = torch.randn_like(k4)
for nq in (1, 2):
    q = torch.randn(b, 32, nq, d, device="cuda", dtype=torch.bfloat16)
    for _ in range(5):
        tree_attn_decode(q, k4, v4, shard_kv_seq=False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
        tree_attn_decode(q, k4, v4, shard_kv_seq=False)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / 50 * 1e6
    print(f"GQA 32q/4kv nq={nq}: {us:8.1f} us/step")
