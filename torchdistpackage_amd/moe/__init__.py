from .layer import (ExpertParallelMoE, TopKRouter, Expert,
                    mark_expert_parallel, is_expert_param)
