"""torchdistpackage_amd — MI355X-native mixed-parallel training toolkit.

A from-scratch rebuild of the capabilities of KimmiShi/TorchDistPackage
(reference at /root/reference, cited per-module) designed for AMD Instinct
MI355X nodes: one process per GPU, RCCL over xGMI via torch.distributed,
hand-written gfx950 HIP kernels for the hot ops (norms, attention, optimizer,
grad utilities), HIP streams for comm/compute overlap.

Public surface mirrors the reference __init__
(/root/reference/torchdistpackage/__init__.py:1-24) plus the MI355X additions
(MoE all-to-all layer, fused ops, model zoo).
"""

__version__ = "0.1.0"

from .dist import (setup_distributed, tpc, torch_parallel_context,
                   is_using_pp, test_comm, setup_node_groups, ShardedEMA,
                   get_mp_ckpt_suffix, mp_ckpt_name, ProcessTopology,
                   hip_prof_start, hip_prof_stop, cu_prof_start, cu_prof_stop,
                   roctx_decorator, nvtx_decorator, ROCTXContext, NVTXContext,
                   has_inf_or_nan, disable_non_master_print, bench_collectives,
                   save_checkpoint, load_checkpoint, latest_step)
from .ddp import (NaiveDdp, NaiveDDP, MoEDP, create_moe_dp_hooks,
                  moe_dp_iter_step, Bf16ZeroOptimizer)
from .utils import fix_rand, partition_by_numel
from .tools import (fix_rand as _fix_rand_alias,  # noqa: F401
                    report_prof, register_profile_hooks, get_model_profile,
                    replace_all_module, register_nan_hooks, check_model_params)
