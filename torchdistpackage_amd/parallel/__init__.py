from .tensor import (TpLinear, ColParallelLinear, RowParallelLinear,
                     Mlp, TpMlp, Attention, TpAttention,
                     Block, ParallelBlock, Transformer,
                     set_tp_group, get_tp_group, get_tp_size, get_tp_rank)
