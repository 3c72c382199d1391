from .tensor import (TpLinear, ColParallelLinear, RowParallelLinear,
                     Mlp, TpMlp, Attention, TpAttention,
                     Block, ParallelBlock, Transformer,
                     set_tp_group, get_tp_group, get_tp_size, get_tp_rank)
from .pipeline import (forward_backward, forward_eval, partition_uniform,
                       partition_balanced, flatten_model, flatten_sequence,
                       flat_and_partition, clip_grad_norm_, NativeScalerPP)
from .context import ulysses_attention, UlyssesAttention, ring_attention
