from .naive_ddp import NaiveDdp, NaiveDDP, GradBucket
from .moe_dp import MoEDP, create_moe_dp_hooks, moe_dp_iter_step
from .zero_optim import Bf16ZeroOptimizer
