#!/bin/bash
# SLURM launcher for one 8-GPU MI355X node (reference run.sh rebuilt:
# the reference requested 1 GPU and ran single-process DataParallel;
# here: one task, torchrun spawning one rank per GPU over RCCL).
#SBATCH --job-name=mgproto
#SBATCH --nodes=1
#SBATCH --ntasks=1
#SBATCH --gpus-per-node=8
#SBATCH --cpus-per-task=64
#SBATCH --mem=256G
#SBATCH --time=2-00:00:00
#SBATCH --output=logs/mgproto.%j.log

export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p logs

srun python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
    --master-addr 127.0.0.1 --master-port 29511 \
    train.py -arch resnet50 -mem_sz 800 -mine_level 20 "$@"
