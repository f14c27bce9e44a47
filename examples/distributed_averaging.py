#!/usr/bin/env python3
"""Standalone distributed averaging with the gossipers (no NN training).

The reference README notes the gossip modules are usable for generic
decentralized averaging (reference README.md:67-68).  Run:

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 \
        --master-addr 127.0.0.1 examples/distributed_averaging.py
"""

import os
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import torch
import torch.distributed as dist

from stochastic_gradient_push_amd import (
    NPeerDynamicDirectedExponentialGraph,
    PushSum,
)


def main():
    dist.init_process_group(
        "nccl" if torch.cuda.is_available() else "gloo"
    )
    rank = dist.get_rank()
    world = dist.get_world_size()
    device = (
        torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
        if torch.cuda.is_available() else torch.device("cpu")
    )
    if device.type == "cuda":
        torch.cuda.set_device(device)

    torch.manual_seed(rank)
    x = torch.randn(1000, device=device)
    target = x.clone()
    dist.all_reduce(target)
    target /= world

    gossiper = PushSum(
        torch.zeros_like(x),
        graph=NPeerDynamicDirectedExponentialGraph(rank, world),
        device=device,
    )
    w = torch.ones(1, device=device)
    for it in range(50):
        x, w = gossiper.mix(x.clone(), w, residual=False)
        x = x.clone()
        w = w.clone()
    err = (x / w - target).abs().max().item()
    print(f"rank {rank}: max error vs true average = {err:.2e}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
