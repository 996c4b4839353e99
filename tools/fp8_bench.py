"""MX-FP8 forward micro-bench vs the bf16 forward at the headline shape."""
import sys, time, torch
sys.path.insert(0, "/root/repo")
from ring_attention_amd.ops.fp8 import quantize_fp8, flash_attn_fp8_quantized
from ring_attention_amd.ops import hip_ext

b, n, h, d = 1, 8192, 8, 64
torch.manual_seed(0)
q = torch.randn(b, n, h, d, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
args = quantize_fp8(q, k, v)
sm = d ** -0.5
def step():
    flash_attn_fp8_quantized(*args, sm)
for _ in range(10): step()
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(50): step()
torch.cuda.synchronize()
us = (time.perf_counter() - t0) / 50 * 1e6
fl = 4 * b * n * n * d * h
print(f"fp8 fwd: {us:8.1f} us  {fl / (us * 1e-6) / 1e12:7.1f} TF")
# quantization cost (one-time per prefill)
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(10): quantize_fp8(q, k, v)
torch.cuda.synchronize()
print(f"quantize: {(time.perf_counter() - t0) / 10 * 1e6:8.1f} us")
# 32k seq
n = 32768
q = torch.randn(b, n, h, d, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
args = quantize_fp8(q, k, v)
for _ in range(3): step()
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(20): step()
torch.cuda.synchronize()
us = (time.perf_counter() - t0) / 20 * 1e6
fl = 4 * b * n * n * d * h
print(f"fp8 fwd 32k: {us:8.1f} us  {fl / (us * 1e-6) / 1e12:7.1f} TF")

# fp8 KV-cache decode vs bf16 decode at 128k / 1M
from ring_attention_amd.ops.fp8 import quantize_kv_cache
from ring_attention_amd.tree_decode import tree_attn_decode, tree_attn_decode_fp8
for n in (131072, 1048576):
    b, h, d = 1, 8, 64
    q = torch.randn(b, h, 1, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, h, n, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, h, n, d, device="cuda", dtype=torch.bfloat16)
    k8, v8, ks, vs = quantize_kv_cache(k, v)
    for name, fn in (("bf16", lambda: tree_attn_decode(q, k, v, shard_kv_seq=False)),
                     ("fp8 ", lambda: tree_attn_decode_fp8(q, k8, v8, ks, vs))):
        for _ in range(5): fn()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(30): fn()
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / 30 * 1e6
        print(f"decode {name} n={n}: {us:8.1f} us/step")
