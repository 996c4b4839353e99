from .attention import RMSNorm, RingAttention
from .rotary import RingRotaryEmbedding, apply_rotary_pos_emb, rotate_half
from .transformer import FeedForward, RingTransformer

__all__ = [
    "RMSNorm", "RingAttention", "RingRotaryEmbedding", "apply_rotary_pos_emb",
    "rotate_half", "FeedForward", "RingTransformer",
]
