import sys, torch
sys.path.insert(0, "/root/repo")
from ring_attention_amd.ops.fp8 import flash_attn_fp8, quantize_fp8
b, n, h, d = 1, 256, 1, 64
torch.manual_seed(7)
q = torch.randn(b, n, h, d, device="cuda", dtype=torch.bfloat16)
k = torch.randn(b, n, h, d, device="cuda", dtype=torch.bfloat16)
v = torch.randn(b, n, h, d, device="cuda", dtype=torch.bfloat16)
out, lse = flash_attn_fp8(q, k, v)
qf, kf, vf = q.float().cpu(), k.float().cpu(), v.float().cpu()
sim = torch.einsum("bihd,bjhd->bhij", qf, kf) * d ** -0.5
ref = torch.einsum("bhij,bjhd->bihd", sim.softmax(-1), vf)
ref_lse = sim.logsumexp(dim=-1)
print("lse err:", (lse.cpu() - ref_lse).abs().max().item())
o = out.float().cpu()
print("out mean rel:", (o-ref).abs().mean().item()/ref.abs().mean().item(),
      "max:", (o-ref).abs().max().item())
# simulate the kernel's P quantization in python: P8 = e4m3(P), out2 = P8 V8
p_t = sim.softmax(-1)
m = sim.max(-1, keepdim=True).values
p_unn = torch.exp(sim - m)  # matches kernel P (max=1)
p8 = p_unn.to(torch.float8_e4m3fn).float()
# v quant per (d, chunk64)
q8_, k8_, v8t_, qs_, ks_, vs_ = quantize_fp8(q, k, v)
v8 = v8t_.view(torch.float8_e4m3fn).float().cpu()        # (b,h,d,n)
vsc = torch.exp2(vs_.float().cpu() - 127).repeat_interleave(64, dim=-1)
vdq = (v8 * vsc)                                          # (b,h,d,n)
num = torch.einsum("bhij,bhdj->bihd", p8, vdq)
den = p8.sum(-1)
sim_ref = num / den.permute(0,2,1).unsqueeze(-1)
print("python-sim fp8P out vs ref:", (sim_ref-ref).abs().mean().item()/ref.abs().mean().item())
print("kernel vs python-sim:", (o-sim_ref).abs().mean().item()/ref.abs().mean().item(),
      "max:", (o-sim_ref).abs().max().item())
