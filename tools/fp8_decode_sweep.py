import sys, time, torch
sys.path.insert(0, "/root/repo")
from ring_attention_amd.ops.fp8 import quantize_kv_cache
from ring_attention_amd.ops import hip_ext
ext = hip_ext.require()
for n in (131072, 1048576):
    b, h, d = 1, 8, 64
    q = torch.randn(b, h, 1, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, h, n, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, h, n, d, device="cuda", dtype=torch.bfloat16)
    k8, v8, ks, vs = quantize_kv_cache(k, v)
    sm = d ** -0.5
    for ch in (0, 32, 64, 128, 256, 512, 1024):
        def fn():
            o, l = ext.decode_partial_fp8(q, k8, v8, ks, vs, sm, ch)
            m = l.max(dim=0).values
            w = (l - m[None]).exp()
            (o * w).sum(dim=0)
        for _ in range(5): fn()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(30): fn()
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / 30 * 1e6
        print(f"n={n} chunks={ch:5d}: {us:8.1f} us")
