from .ulysses import ulysses_attention, UlyssesAttention
from .ring import ring_attention
