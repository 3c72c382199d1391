#!/bin/bash
# 2-node / 16-GPU template (reference parity: /root/reference/tools/sbatch.sh)
# runs the collective bandwidth benchmark via srun; adapt the payload line.
#SBATCH -J tdpa-comm
#SBATCH -N 2
#SBATCH --ntasks-per-node=8
#SBATCH --gpus-per-node=8
#SBATCH -t 00:30:00
export HSA_ENABLE_IPC_MODE_LEGACY=0
srun python -m torchdistpackage_amd.dist.comm_bench
