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
