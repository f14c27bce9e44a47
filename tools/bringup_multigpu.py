#!/usr/bin/env python3
"""Multi-GPU RCCL bring-up smoke (first thing to run on an N-GPU node).

Launch:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 --master-port 29631 \
        tools/bringup_multigpu.py

Checks, in order (each prints PASS/FAIL on rank 0):
  1. RCCL all-reduce sanity on the default world.
  2. Grouped p2p gossip exchange (PushSum.mix) over c10d -> consensus.
  3. The native RcclTransport (comm_backend='rccl') exchange -> same.
  4. GossipDataParallel zero-lr consensus with bf16 wire + master
     weights (the bench configuration), 20 steps.
"""

import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def log(rank, msg):
    if rank == 0:
        print(f"[bringup] {msg}", flush=True)


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local = int(os.environ.get("LOCAL_RANK", rank))
    torch.cuda.set_device(local)
    dist.init_process_group("nccl")

    # 1. all-reduce sanity
    t = torch.ones(1 << 20, device="cuda") * (rank + 1)
    dist.all_reduce(t)
    expect = world * (world + 1) / 2
    assert torch.allclose(t, torch.full_like(t, expect)), t[0]
    log(rank, f"1. RCCL all-reduce over {world} ranks: PASS")

    # 2. gossip consensus over c10d grouped p2p
    from stochastic_gradient_push_amd import gossiper as G
    from stochastic_gradient_push_amd.graphs import (
        NPeerDynamicDirectedExponentialGraph,
    )

    def consensus(transport=None):
        torch.manual_seed(rank)
        graph = NPeerDynamicDirectedExponentialGraph(rank, world)
        x = torch.randn(1 << 20, device="cuda")
        w = torch.ones(1, device="cuda")
        target = x.clone()
        dist.all_reduce(target)
        target /= world
        gsp = G.PushSum(
            torch.zeros_like(x), graph=graph, device=x.device,
            rank=rank, world_size=world, transport=transport,
        )
        for _ in range(60):
            x, w = gsp.mix(x.clone(), w.clone(), residual=False)
            x = x.clone()
            w = w.clone().reshape(1)
        est = x / w
        err = (est - target).abs().max().item()
        assert err < 1e-3, err
        return err

    err = consensus()
    log(rank, f"2. c10d grouped-p2p push-sum consensus: PASS (err {err:.1e})")

    # 3. native transport
    from stochastic_gradient_push_amd.comm import create_rccl_transport

    transport = create_rccl_transport()
    err = consensus(transport=transport)
    log(rank, f"3. native RcclTransport consensus: PASS (err {err:.1e})")

    # 4. wrapper end-to-end at the bench configuration
    import torch.nn as nn

    from stochastic_gradient_push_amd import GossipDataParallel
    from stochastic_gradient_push_amd.models import resnet18
    from stochastic_gradient_push_amd.ops.fused_sgd import FusedSGD

    torch.manual_seed(rank)
    model = resnet18(num_classes=10).cuda().to(
        memory_format=torch.channels_last
    )
    gdp = GossipDataParallel(
        model, push_sum=True, gossip_dtype=torch.bfloat16,
        working_dtype=torch.bfloat16,
    )
    target = gdp.flatp.flat.clone()
    dist.all_reduce(target)
    target /= world
    opt = FusedSGD(gdp.flatp, lr=0.0)
    x = torch.randn(4, 3, 32, 32, device="cuda").contiguous(
        memory_format=torch.channels_last
    )
    y = torch.randint(0, 10, (4,), device="cuda")
    gdp.train()
    for _ in range(25):
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            loss = nn.functional.cross_entropy(gdp(x), y)
        loss.backward()
        opt.step()
        opt.zero_grad()
        gdp.transfer_params()
    gdp.sync_comms()
    gdp._query_gossip_queue(non_blocking=False)
    gdp.unbias()
    err = (gdp.flatp.flat - target).abs().max().item()
    assert err < 3e-2, err  # bf16 wire resolution
    log(rank, f"4. GossipDataParallel (bf16 wire + master weights) "
              f"consensus: PASS (err {err:.1e})")

    gdp.shutdown()
    dist.barrier()
    dist.destroy_process_group()
    log(rank, "ALL BRING-UP CHECKS PASSED")


if __name__ == "__main__":
    main()
