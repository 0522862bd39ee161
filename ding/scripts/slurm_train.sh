#!/usr/bin/env bash
#SBATCH --job-name=ding-train
#SBATCH --nodes=1
#SBATCH --ntasks-per-node=8
#SBATCH --gpus-per-node=8
# One process per MI355X over RCCL; the SLURM parser fills per-rank args.
# Usage: sbatch slurm_train.sh <config.py>
CFG=${1:?usage: sbatch slurm_train.sh <config.py>}
export HSA_ENABLE_IPC_MODE_LEGACY=0
srun python -m torch.distributed.run --nnodes="$SLURM_NNODES" \
  --nproc-per-node=8 --master-addr 127.0.0.1 \
  -m ding.entry.cli -m serial -c "$CFG" -s 0
