from .reference import default_attention, softclamp, MASK_VALUE
from .ring_flash import (
    RingFlashAttentionFunction,
    ring_flash_attn,
    ring_flash_attn_,
)

__all__ = [
    "default_attention", "softclamp", "MASK_VALUE",
    "RingFlashAttentionFunction", "ring_flash_attn", "ring_flash_attn_",
]
