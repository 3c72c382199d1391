from .tp_utils import (set_tp_group, get_tp_group, get_tp_size, get_tp_rank,
                       TpLinear, ColParallelLinear, RowParallelLinear,
                       copy_to_tp_region, reduce_from_tp_region,
                       gather_from_sequence_parallel_region,
                       reduce_scatter_to_sequence_parallel_region,
                       maybe_gather_for_sequence_parallel,
                       maybe_split_into_sequence_parallel,
                       set_sequence_parallel_attr, is_sequence_parallel,
                       allreduce_sequence_parallel_grads,
                       mark_sequence_parallel_params)
from .mlp import Mlp, TpMlp
from .attn import Attention, TpAttention
from .transformer import Block, ParallelBlock, Transformer
