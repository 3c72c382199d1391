from .schedule import forward_backward, forward_eval
from .partition import (partition_uniform, partition_balanced, partition_by_time, flatten_model,
                        flatten_sequence, flat_and_partition, CallableModule)
from .grad_clip import clip_grad_norm_, NativeScalerPP
from . import p2p
