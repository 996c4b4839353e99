"""Per-tile segment timing of the v2 forward kernel (s_memtime stamps).
7 stamps per tile: [0 A-entry][1 QKT end][2 finish end][3 B-entry]
[4 PV end][5 softmax end][6 tile end].  Run with RING_ATTN_FWD_V2=1."""
import os, sys, torch
sys.path.insert(0, "/root/repo")
os.environ.setdefault("RING_ATTN_FWD_V2", "1")
from ring_attention_amd.ops import hip_ext
ext = hip_ext.require()
b, n, h, d = 1, 8192, 8, 64
torch.manual_seed(0)
q = torch.randn(b, n, h, d, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
out = torch.empty_like(q)
lse = torch.empty(b, h, n, device="cuda", dtype=torch.float32)
ticks = torch.zeros(64 * 7, device="cuda", dtype=torch.int64)
scale = d ** -0.5
for _ in range(3):
    ext.attn_fwd(q, k, v, None, None, None, None, out, lse, scale,
                 False, 0, 1, 0, False, False, 50.0, True, True, 1, 0, ticks)
torch.cuda.synchronize()
t = ticks.view(64, 7).cpu().numpy()
import numpy as np
names = ["A:QKT", "A:finish", "A:bar", "B:PV+tr", "B:dma+sm", "B:wait", "barrier"]
d0 = np.diff(t, axis=1).astype(float)
tile_total = t[1:, 0] - t[:-1, 0]
print("median cycles per tile segment (tiles 4..60):")
for i, nm in enumerate(names[:6]):
    print(f"  {nm:10s} {np.median(d0[4:60, i]):8.0f}")
print(f"  whole tile (t->t): {np.median(tile_total[4:60]):8.0f}")
