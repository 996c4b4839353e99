"""CPU coverage for the MX-FP8 wrapper/ring logic.

The GPU kernels are GPU-marked in test_gpu_kernels.py; here the eager
(dequantized) fallback exercises the quantization, the padding/slicing,
the masking semantics, and — through gloo — the ring transport carrying
uint8 shard bundles (a wire path the bf16 suites never touch)."""

import pytest
import torch

from ring_attention_amd.ops.fp8 import flash_attn_fp8, ring_flash_attn_fp8

from .distributed_utils import run_distributed


@pytest.mark.parametrize("n,causal,groups", [(300, False, 1), (512, True, 2),
                                             (777, True, 1)])
def test_fp8_eager_fallback(n, causal, groups):
    b, h, d = 1, 4, 64
    hk = h // groups
    torch.manual_seed(5)
    q = torch.randn(b, n, h, d, dtype=torch.bfloat16)
    k = torch.randn(b, n, hk, d, dtype=torch.bfloat16)
    v = torch.randn(b, n, hk, d, dtype=torch.bfloat16)
    out, lse = flash_attn_fp8(q, k, v, causal=causal)
    assert out.shape == (b, n, h, d) and lse.shape == (b, h, n)
    kf = k.float().repeat(1, 1, groups, 1)
    vf = v.float().repeat(1, 1, groups, 1)
    sim = torch.einsum("bihd,bjhd->bhij", q.float(), kf) * d ** -0.5
    if causal:
        pos = torch.arange(n)
        sim = sim.masked_fill((pos[None, :] > pos[:, None])[None, None],
                              float("-inf"))
    ref = torch.einsum("bhij,bjhd->bihd", sim.softmax(-1), vf)
    rel = ((out.float() - ref).abs().mean() / ref.abs().mean()).item()
    assert rel < 0.06, f"fp8 eager rel {rel}"
    # causal short rows: few keys -> q/k quantization error shows directly
    assert (lse - sim.logsumexp(-1)).abs().max().item() < 0.09


def _ring_case(rank, world, causal):
    b, n, h, d = 1, 512, 2, 64
    torch.manual_seed(19)
    q = torch.randn(b, n, h, d, dtype=torch.bfloat16)
    k = torch.randn(b, n, h, d, dtype=torch.bfloat16)
    v = torch.randn(b, n, h, d, dtype=torch.bfloat16)
    ref, ref_lse = flash_attn_fp8(q, k, v, causal=causal)

    qs = q.chunk(world, dim=1)[rank]
    ks = k.chunk(world, dim=1)[rank]
    vs = v.chunk(world, dim=1)[rank]
    out, lse = ring_flash_attn_fp8(qs, ks, vs, causal=causal)

    want = ref.chunk(world, dim=1)[rank]
    e = (out.float() - want.float()).abs().max().item()
    s = want.float().abs().max().item() + 1e-6
    assert e / s < 8e-2, f"rank {rank} rel {e/s}"
    want_lse = ref_lse.chunk(world, dim=2)[rank]
    assert (lse - want_lse).abs().max().item() < 0.15
    return True


@pytest.mark.parametrize("world,causal", [(2, False), (2, True), (4, True)])
def test_fp8_ring_gloo(world, causal):
    # uint8 shard bundles over the real gloo ring transport
    run_distributed(world, _ring_case, causal)


def _decode_case(rank, world):
    from ring_attention_amd.ops.fp8 import quantize_kv_cache
    from ring_attention_amd.tree_decode import tree_attn_decode, tree_attn_decode_fp8
    b, h, n, d = 1, 4, 256, 64
    torch.manual_seed(37)
    q = torch.randn(b, h, 1, d)
    k = torch.randn(b, h, n, d)
    v = torch.randn(b, h, n, d)
    ref = tree_attn_decode(q, k, v, shard_kv_seq=False) if world == 1 else None
    ks_ = k.chunk(world, dim=-2)[rank]
    vs_ = v.chunk(world, dim=-2)[rank]
    cache = quantize_kv_cache(ks_, vs_)
    out = tree_attn_decode_fp8(q, *cache)
    return out


def test_fp8_decode_cpu_single():
    from ring_attention_amd.ops.fp8 import quantize_kv_cache
    from ring_attention_amd.tree_decode import tree_attn_decode, tree_attn_decode_fp8
    b, h, n, d = 2, 4, 512, 64
    torch.manual_seed(37)
    q = torch.randn(b, h, 1, d)
    k = torch.randn(b, h, n, d)
    v = torch.randn(b, h, n, d)
    ref = tree_attn_decode(q, k, v, shard_kv_seq=False)
    out = tree_attn_decode_fp8(q, *quantize_kv_cache(k, v))
    rel = ((out.float() - ref.float()).abs().max()
           / (ref.float().abs().max() + 1e-6)).item()
    assert rel < 5e-2, f"fp8 decode cpu rel {rel}"


def test_fp8_decode_gloo_w2():
    # sharded fp8 cache + the 2-round collective merge, CPU fallback partial
    results = run_distributed(2, _decode_case)
    a, b_ = results
    assert (a - b_).abs().max().item() < 1e-6  # all ranks return the merged out


@pytest.mark.parametrize("d", [32, 40, 96])
def test_fp8_head_dim_padding(d):
    # any head dim <= 128 via exact zero-pad (bf16-path parity)
    b, n, h = 1, 256, 2
    torch.manual_seed(43)
    q = torch.randn(b, n, h, d, dtype=torch.bfloat16)
    k = torch.randn(b, n, h, d, dtype=torch.bfloat16)
    v = torch.randn(b, n, h, d, dtype=torch.bfloat16)
    out, lse = flash_attn_fp8(q, k, v)
    assert out.shape == (b, n, h, d)
    sim = torch.einsum("bihd,bjhd->bhij", q.float(), k.float()) * d ** -0.5
    ref = torch.einsum("bhij,bjhd->bihd", sim.softmax(-1), v.float())
    rel = ((out.float() - ref).abs().mean() / ref.abs().mean()).item()
    assert rel < 0.07, f"fp8 d={d} rel {rel}"


def test_ring_attention_module_fp8_inference():
    # L3 integration: fp8_inference routes no-grad forwards to the e4m3
    # path (eager fallback on CPU), training steps keep the bf16/fp32 path
    from ring_attention_amd.models.attention import RingAttention
    torch.manual_seed(47)
    attn = RingAttention(dim=128, dim_head=64, heads=2, causal=True,
                         fp8_inference=True).to(torch.bfloat16)
    x = torch.randn(1, 256, 128, dtype=torch.bfloat16)
    with torch.no_grad():
        out_fp8 = attn(x)
    attn.fp8_inference = False
    with torch.no_grad():
        ref = attn(x)
    rel = ((out_fp8.float() - ref.float()).abs().mean()
           / (ref.float().abs().mean() + 1e-9)).item()
    assert out_fp8.shape == ref.shape
    assert rel < 0.2, f"fp8 module path rel {rel}"   # quantization floor
    # with grad enabled the fp8 path must NOT engage (it has no backward)
    x2 = torch.randn(1, 256, 128, dtype=torch.bfloat16, requires_grad=True)
    out = attn_train = RingAttention(dim=128, dim_head=64, heads=2, causal=True,
                                     fp8_inference=True).to(torch.bfloat16)(x2)
    out.sum().backward()
    assert x2.grad is not None


def test_ring_transformer_fp8_inference():
    # L4 passthrough: fp8_inference reaches every layer's attention; logits
    # track the bf16 model closely (attention quantization washes through
    # the projections)
    from ring_attention_amd import RingTransformer
    torch.manual_seed(3)
    kw = dict(num_tokens=64, dim=128, depth=2, causal=True, dim_head=64,
              heads=2, bucket_size=256, ring_seq_size=512)
    m = RingTransformer(**kw).bfloat16()
    m8 = RingTransformer(**kw, fp8_inference=True).bfloat16()
    m8.load_state_dict(m.state_dict())
    ids = torch.randint(0, 64, (1, 256))
    with torch.no_grad():
        ref = m(ids)
        out = m8(ids)
    rel = ((out.float() - ref.float()).abs().mean()
           / (ref.float().abs().mean() + 1e-9)).item()
    assert rel < 0.05, f"transformer fp8 rel {rel}"


@pytest.mark.parametrize("nq,nk", [(64, 64), (256, 1024), (128, 2000)])
def test_fp8_cross_length(nq, nk):
    # nq != nk (chunked prefill / cross-attention shapes) and tiny lengths
    # far below one tile — the wrapper pads each side independently
    b, h, d = 1, 2, 64
    torch.manual_seed(53)
    q = torch.randn(b, nq, h, d, dtype=torch.bfloat16)
    k = torch.randn(b, nk, h, d, dtype=torch.bfloat16)
    v = torch.randn(b, nk, h, d, dtype=torch.bfloat16)
    out, lse = flash_attn_fp8(q, k, v)
    assert out.shape == (b, nq, h, d) and lse.shape == (b, h, nq)
    sim = torch.einsum("bihd,bjhd->bhij", q.float(), k.float()) * d ** -0.5
    ref = torch.einsum("bhij,bjhd->bihd", sim.softmax(-1), v.float())
    rel = ((out.float() - ref).abs().mean() / ref.abs().mean()).item()
    assert rel < 0.07, f"fp8 nq={nq} nk={nk} rel {rel}"
