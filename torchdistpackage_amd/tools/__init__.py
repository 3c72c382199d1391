from ..utils import fix_rand
from .profiler import (register_profile_hooks, remove_profile_hooks,
                       report_prof, get_model_profile)
from .debug_nan import register_nan_hooks, check_model_params
from .module_replace import replace_all_module

# CUDA-only int8 libraries: optional, like the reference
# (/root/reference/torchdistpackage/__init__.py:19-24)
try:
    from .module_replace import replace_linear_by_bnb  # noqa: F401
except ImportError:
    pass
try:
    from .module_replace import replace_linear_by_bminf  # noqa: F401
except ImportError:
    pass
from .slurm_monitor import monitor_job, submit_job, job_state
