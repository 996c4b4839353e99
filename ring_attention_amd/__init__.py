"""ring_attention_amd — MI355X-native ring attention framework.

Brand-new implementation of the capabilities of lucidrains/ring-attention-pytorch
(reference mounted at /root/reference), designed MI355X-first: PyTorch-ROCm
front end, hand-written CDNA4 (gfx950) HIP flash-attention kernels, RCCL over
xGMI for the ring transport with comm/compute overlap.
"""

from .ops import default_attention, ring_flash_attn, ring_flash_attn_

__version__ = "0.1.0"

__all__ = [
    "default_attention",
    "ring_flash_attn",
    "ring_flash_attn_",
]
