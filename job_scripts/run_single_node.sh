#!/bin/bash
# Single 8-GPU MI355X node via torchrun: one process per GPU over
# RCCL/xGMI (the MI355X-native launch mode; no SLURM needed).
# Usage: ./run_single_node.sh [sgp|osgp|dpsgd|ar] [ngpus]
ALGO=${1:-sgp}
NGPUS=${2:-8}
export HSA_ENABLE_IPC_MODE_LEGACY=0

case $ALGO in
  sgp)   FLAGS="--push_sum True --all_reduce False --graph_type 5" ;;
  osgp)  FLAGS="--push_sum True --all_reduce False --graph_type 5 --overlap True" ;;
  dpsgd) FLAGS="--push_sum False --all_reduce False --graph_type 1" ;;
  ar)    FLAGS="--all_reduce True --graph_type -1" ;;
  *) echo "unknown algo $ALGO"; exit 1 ;;
esac

python -m torch.distributed.run --nnodes=1 --nproc-per-node $NGPUS \
    --master-addr 127.0.0.1 --master-port 40100 \
    gossip_sgd.py \
    --batch_size 32 --lr 0.1 --num_dataloader_workers 8 \
    --num_epochs 90 --nesterov True --warmup True \
    --schedule 30 0.1 60 0.1 80 0.1 \
    --tag "${ALGO}_n${NGPUS}_" --print_freq 100 --verbose False \
    --seed 1 --checkpoint_dir ./checkpoints/ \
    --network_interface_type auto $FLAGS
