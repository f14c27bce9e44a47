#!/bin/bash
#SBATCH --job-name=DPSGD_IB
#SBATCH --output=DPSGD_IB.out
#SBATCH --error=DPSGD_IB.err
#SBATCH --nodes=NB_NODES
#SBATCH --cpus-per-task=32
#SBATCH --gres=gpu:8
#SBATCH --time=30:00:00
#SBATCH --signal=B:USR1@120

# Replace NB_NODES with the number of nodes to use.
# One SLURM task per node; per-node batch 256 (reference recipe).
# RCCL honors NCCL_* env; the trainer pins the fabric via
# --network_interface_type.

export HSA_ENABLE_IPC_MODE_LEGACY=0

srun python -u gossip_sgd.py \
    --batch_size 256 --lr 0.1 --num_dataloader_workers 16 \
    --num_epochs 90 --nesterov True --warmup True \
    --schedule 30 0.1 60 0.1 80 0.1 \
    --train_fast False --master_port 40100 \
    --tag 'DPSGD_IB' --print_freq 100 --verbose False \
    --seed 1 --checkpoint_dir ./checkpoints/ \
    --dataset imagefolder --dataset_dir $IMAGENET_DIR \
    --network_interface_type 'infiniband' --push_sum False --all_reduce False --graph_type 1
