"""Concurrency-protocol tests: interrupted-gossip sentinel, heartbeat,
and the multi-process intra-node tier (nprocs_per_node > 1)."""

import threading

import pytest
import torch
import torch.distributed as dist
import torch.nn as nn

from tests.dist_utils import run_dist


def tiny_model(seed):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(6, 8), nn.ReLU(), nn.Linear(8, 4))


def _sentinel_retry(rank, world_size):
    """A RuntimeError inside the gossip thread must set the ps_weight=-1
    sentinel, and the train thread must re-arm and retry without
    crashing (reference distributed.py:359-364, 500-504)."""
    from stochastic_gradient_push_amd import GossipDataParallel

    gdp = GossipDataParallel(tiny_model(rank), push_sum=True)
    opt = torch.optim.SGD(gdp.parameters(), lr=0.01)
    x = torch.randn(2, 6)
    y = torch.randn(2, 4)
    gdp.train()

    def step():
        loss = ((gdp(x) - y) ** 2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        gdp.transfer_params()

    step()  # normal gossip round so both ranks are in sync
    gdp.sync_comms()

    # inject a failure into the next mix() on every rank
    gossiper = list(gdp.dist_config["gossipers"].values())[0]
    original_mix = gossiper.mix
    fail_once = {"armed": True}

    def failing_mix(*a, **kw):
        if fail_once["armed"]:
            fail_once["armed"] = False
            raise RuntimeError("injected gossip failure")
        return original_mix(*a, **kw)

    with gdp.gossip_lock:
        gossiper.mix = failing_mix

    step()          # this gossip round fails in the background thread
    # the pre-forward query must see the sentinel, re-arm, and carry on
    step()
    step()
    gdp.sync_comms()
    gdp.unbias()
    assert torch.isfinite(gdp.flatp.flat).all()
    # gossip recovered: a fresh round completed after the failure
    assert not fail_once["armed"]
    gdp.shutdown()


def test_interrupted_gossip_sentinel():
    run_dist(_sentinel_retry, world_size=2)


def _nprocs_tier(rank, world_size):
    """2 processes forming ONE node (nprocs_per_node=2): params broadcast
    from the local master each forward, grads averaged across the node
    after backward, gossip skipped on the non-master (reference
    distributed.py:62-78, 278-296, 551-562)."""
    from stochastic_gradient_push_amd import GossipDataParallel

    model = tiny_model(seed=rank)  # deliberately different init
    gdp = GossipDataParallel(
        model, push_sum=True, nprocs_per_node=2, verbose=False,
    )
    opt = torch.optim.SGD(gdp.parameters(), lr=0.05)
    gdp.train()

    torch.manual_seed(100 + rank)  # different data per local process
    x = torch.randn(2, 6)
    y = torch.randn(2, 4)

    for _ in range(3):
        out = gdp(x)  # forward pre-sync: both procs now share params
        loss = ((out - y) ** 2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        gdp.transfer_params()

    # one more forward to broadcast the post-step params, then compare
    gdp(x)
    flat = gdp.flatp.flat.detach().clone()
    gathered = [torch.zeros_like(flat) for _ in range(world_size)]
    dist.all_gather(gathered, flat)
    assert torch.allclose(gathered[0], gathered[1], atol=1e-6), (
        (gathered[0] - gathered[1]).abs().max()
    )


def test_nprocs_per_node_tier():
    run_dist(_nprocs_tier, world_size=2)


def _heartbeat(rank, world_size):
    """Wedged gossip (peer never transfers) must raise after the
    heartbeat timeout rather than hanging forever (reference
    distributed.py:36, 349-352)."""
    from stochastic_gradient_push_amd import distributed as D

    old = D.HEARTBEAT_TIMEOUT
    D.HEARTBEAT_TIMEOUT = 2
    try:
        gdp = D.GossipDataParallel(tiny_model(rank), push_sum=True)
        if rank == 0:
            # rank 0 transfers and then queries; rank 1 never transfers,
            # so rank 0's gossip exchange cannot complete
            gdp.transfer_params()
            with pytest.raises(RuntimeError, match="timeout"):
                gdp._query_gossip_queue(non_blocking=False)
        # release the wedge so shutdown is clean
        dist.barrier()
        if rank == 1:
            gdp.transfer_params()
            gdp.sync_comms()
        else:
            gdp.sync_comms()
    finally:
        D.HEARTBEAT_TIMEOUT = old


def test_gossip_heartbeat_timeout():
    run_dist(_heartbeat, world_size=2)


def _runtime_ppi_change(rank, world_size):
    """update_gossiper('peers_per_itr', v) under the gossip lock changes
    the active out-degree mid-training (reference distributed.py:197-207,
    driven by the trainer's peers-per-itr schedule)."""
    from stochastic_gradient_push_amd import (
        GossipDataParallel,
        NPeerDynamicDirectedExponentialGraph,
    )

    model = tiny_model(seed=rank)
    graph = NPeerDynamicDirectedExponentialGraph(
        rank, world_size, peers_per_itr=2
    )
    gdp = GossipDataParallel(model, graph=graph, push_sum=True)
    opt = torch.optim.SGD(gdp.parameters(), lr=0.0)
    x = torch.randn(2, 6)
    y = torch.randn(2, 4)
    gdp.train()

    def step():
        loss = ((gdp(x) - y) ** 2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        gdp.transfer_params()

    for _ in range(3):
        step()
    gdp.sync_comms()
    gdp.update_gossiper("peers_per_itr", 1)
    gossiper = list(gdp.dist_config["gossipers"].values())[0]
    assert gossiper.peers_per_itr == 1
    for _ in range(3):
        step()
    gdp.sync_comms()
    gdp.unbias()
    assert torch.isfinite(gdp.flatp.flat).all()
    gdp.shutdown()


def test_runtime_peers_per_itr_change():
    run_dist(_runtime_ppi_change, world_size=4)


def _chaos_training(rank, world_size):
    """Stress: nonzero-lr training while gossip randomly fails; training
    must stay finite and keep making progress (sentinel retry path under
    load)."""
    import random

    from stochastic_gradient_push_amd import GossipDataParallel

    torch.manual_seed(rank)
    gdp = GossipDataParallel(tiny_model(rank), push_sum=True)
    opt = torch.optim.SGD(gdp.parameters(), lr=0.02)

    gossiper = list(gdp.dist_config["gossipers"].values())[0]
    original_mix = gossiper.mix
    rng = random.Random(1234)  # same schedule on every rank

    def flaky_mix(*a, **kw):
        if rng.random() < 0.3:
            raise RuntimeError("chaos")
        return original_mix(*a, **kw)

    with gdp.gossip_lock:
        gossiper.mix = flaky_mix

    torch.manual_seed(99)
    w_true = torch.randn(6, 4)
    torch.manual_seed(200 + rank)
    x = torch.randn(16, 6)
    y = x @ w_true

    gdp.train()
    first = last = None
    for _ in range(25):
        loss = ((gdp(x) - y) ** 2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        gdp.transfer_params()
        if first is None:
            first = loss.item()
        last = loss.item()
    gdp.sync_comms()
    gdp.unbias()
    assert torch.isfinite(gdp.flatp.flat).all()
    assert last < first  # still optimizing through the failures
    gdp.shutdown()


def test_chaos_gossip_failures():
    run_dist(_chaos_training, world_size=2)


def _asynch_mode(rank, world_size):
    """synch_freq > 0 (reference distributed.py:336-387): the forward
    hook polls gossip NON-blocking for up to synch_freq iterations
    before forcing a blocking wait; asynch mode disables lazy mixing
    (explicit bias/de-bias) and must still reach consensus at zero lr."""
    import torch
    import torch.distributed as dist
    import torch.nn as nn

    from stochastic_gradient_push_amd import GossipDataParallel

    torch.manual_seed(rank)
    model = nn.Sequential(nn.Linear(6, 12), nn.ReLU(), nn.Linear(12, 3))
    target = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    dist.all_reduce(target)
    target /= world_size

    gdp = GossipDataParallel(model, push_sum=True, synch_freq=3)
    assert gdp.asynch and not gdp.lazy_mixing
    opt = torch.optim.SGD(gdp.parameters(), lr=0.0)
    x = torch.randn(4, 6)
    y = torch.randn(4, 3)
    gdp.train()
    for _ in range(60):
        loss = ((gdp(x) - y) ** 2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        gdp.transfer_params()
    gdp.sync_comms()
    gdp._query_gossip_queue(non_blocking=False)
    gdp.unbias()
    flat = gdp.flatp.flat.detach()
    assert torch.allclose(flat, target, atol=2e-3), (
        f"rank {rank}: max err {(flat - target).abs().max()}"
    )
    gdp.shutdown()


def test_asynch_synch_freq_consensus():
    from tests.dist_utils import run_dist

    run_dist(_asynch_mode, world_size=2)
