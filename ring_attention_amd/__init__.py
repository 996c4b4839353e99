"""ring_attention_amd — MI355X-native ring attention framework.

Brand-new implementation of the capabilities of lucidrains/ring-attention-pytorch
(reference mounted at /root/reference), designed MI355X-first: PyTorch-ROCm
front end, hand-written CDNA4 (gfx950) HIP flash-attention kernels, RCCL over
xGMI for the ring transport with comm/compute overlap.

Public API parity with the reference package __init__
(/root/reference/ring_attention_pytorch/__init__.py:1-21); the reference's
``ring_flash_attn_cuda`` has the HIP-native counterpart ``ring_flash_attn_hip``.
"""

from .ops import default_attention, ring_flash_attn, ring_flash_attn_
from .models import (
    FeedForward,
    RMSNorm,
    RingAttention,
    RingRotaryEmbedding,
    RingTransformer,
    apply_rotary_pos_emb,
)
from .tree_decode import tree_attn_decode, tree_attn_decode_fp8
from .zigzag import zig_zag_attn, zig_zag_pad_seq, zig_zag_shard

__version__ = "0.1.0"

__all__ = [
    "default_attention",
    "ring_flash_attn",
    "ring_flash_attn_",
    "ring_flash_attn_hip",
    "ring_flash_attn_hip_",
    "flash_attn",
    "flash_attn_offset",
    "flash_attn_fp8",
    "ring_flash_attn_fp8",
    "quantize_fp8",
    "quantize_kv_cache",
    "RingAttention",
    "RingTransformer",
    "RingRotaryEmbedding",
    "apply_rotary_pos_emb",
    "RMSNorm",
    "FeedForward",
    "tree_attn_decode",
    "tree_attn_decode_fp8",
    "zig_zag_attn",
    "zig_zag_pad_seq",
    "zig_zag_shard",
]


def __getattr__(name):
    # HIP-backed function imports lazily (requires the built extension on GPU)
    if name in ("ring_flash_attn_hip", "ring_flash_attn_hip_", "flash_attn",
                "flash_attn_offset"):
        from .ops import ring_flash_hip
        return getattr(ring_flash_hip, name)
    if name in ("flash_attn_fp8", "ring_flash_attn_fp8", "quantize_fp8",
                "quantize_kv_cache"):
        from .ops import fp8
        return getattr(fp8, name)
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
