from .launch import setup_distributed, find_free_port
from .topo import ProcessTopology, tpc, torch_parallel_context, is_using_pp, \
    test_comm, gen_axis_groups
from .node_group import setup_node_groups, get_node_group, reset_node_groups
from .sharded_ema import ShardedEMA, partition_by_numel
from .mp_ckpt import get_mp_ckpt_suffix, mp_ckpt_name
from .utils import (hip_prof_start, hip_prof_stop, cu_prof_start, cu_prof_stop,
                    roctx_decorator, nvtx_decorator, ROCTXContext, NVTXContext,
                    has_inf_or_nan, disable_non_master_print, restore_print)
from .comm_bench import bench_collectives
from .checkpoint import save_checkpoint, load_checkpoint, latest_step
