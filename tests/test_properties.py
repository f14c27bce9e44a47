"""Property-based topology tests (hypothesis) + algorithm-level
convergence: gossip SGD actually optimizes, not just reaches consensus."""

import pytest
import torch
import torch.distributed as dist
import torch.nn as nn
from hypothesis import given, settings
from hypothesis import strategies as st

from stochastic_gradient_push_amd.graphs import (
    DynamicBipartiteExponentialGraph,
    DynamicDirectedExponentialGraph,
    NPeerDynamicDirectedExponentialGraph,
    RingGraph,
)
from tests.dist_utils import run_dist

DIRECTED = [
    DynamicDirectedExponentialGraph,
    NPeerDynamicDirectedExponentialGraph,
    RingGraph,
]


@settings(max_examples=40, deadline=None)
@given(
    world_size=st.integers(min_value=2, max_value=48),
    cls_idx=st.integers(min_value=0, max_value=len(DIRECTED) - 1),
    steps=st.integers(min_value=1, max_value=6),
)
def test_directed_graphs_always_consistent(world_size, cls_idx, steps):
    """For any world size: every send has exactly one matching receive at
    every rotation step, out-degree == peers_per_itr, no self-edges."""
    cls = DIRECTED[cls_idx]
    graphs = [cls(r, world_size) for r in range(world_size)]
    n_steps = steps if graphs[0].is_dynamic_graph() else 1
    for step in range(n_steps):
        rotate = step > 0
        outs, ins = set(), set()
        for g in graphs:
            o, i = g.get_edges(rotate=rotate)
            assert len(o) == g.peers_per_itr
            for e in o:
                assert e.src != e.dest
            outs.update((e.src, e.dest) for e in o)
            ins.update((e.src, e.dest) for e in i)
        assert outs == ins


@settings(max_examples=20, deadline=None)
@given(half=st.integers(min_value=1, max_value=16))
def test_bipartite_even_world_consistent(half):
    """Bipartite graphs (even world sizes, as the reference assumes):
    edges always cross parity and sends match receives."""
    world_size = 2 * half
    if world_size < 2:
        return
    graphs = [
        DynamicBipartiteExponentialGraph(r, world_size)
        for r in range(world_size)
    ]
    for step in range(3):
        rotate = step > 0
        outs, ins = set(), set()
        for g in graphs:
            o, i = g.get_edges(rotate=rotate)
            for e in o:
                assert (e.src % 2) != (e.dest % 2)
            outs.update((e.src, e.dest) for e in o)
            ins.update((e.src, e.dest) for e in i)
        assert outs == ins


@settings(max_examples=30, deadline=None)
@given(
    world_size=st.integers(min_value=2, max_value=64),
    ppi=st.integers(min_value=1, max_value=3),
)
def test_npeer_rotation_covers_phone_book(world_size, ppi):
    if ppi >= world_size:
        return
    g = NPeerDynamicDirectedExponentialGraph(0, world_size, peers_per_itr=ppi)
    book = len(g.phone_book[0])
    if book < ppi:
        return  # tiny worlds where degree < requested ppi
    seen = set()
    for _ in range(book * 2):
        out, _ = g.get_edges(rotate=True)
        seen.update(e.dest for e in out)
    assert seen == {e.dest for e in g.phone_book[0]}


# ------------------------------------------------------- convergence


def _sgp_actually_optimizes(rank, world_size):
    """2-rank SGP on a fixed least-squares problem: the training loss
    must drop by >80% — gossip training optimizes, not just averages."""
    from stochastic_gradient_push_amd import GossipDataParallel

    torch.manual_seed(10 + rank)
    model = nn.Linear(8, 1)
    gdp = GossipDataParallel(model, push_sum=True)
    opt = torch.optim.SGD(gdp.parameters(), lr=0.05)

    # distinct data shards with a shared true solution
    torch.manual_seed(99)
    w_true = torch.randn(8, 1)
    torch.manual_seed(500 + rank)
    x = torch.randn(64, 8)
    y = x @ w_true

    gdp.train()
    losses = []
    for _ in range(60):
        loss = ((gdp(x) - y) ** 2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        gdp.transfer_params()
        losses.append(loss.item())
    gdp.sync_comms()
    gdp.unbias()
    assert losses[-1] < 0.2 * losses[0], (losses[0], losses[-1])
    gdp.shutdown()


def test_sgp_training_converges():
    run_dist(_sgp_actually_optimizes, world_size=2)


def test_gossip_alias_package():
    import gossip
    import gossip.utils

    assert gossip.GossipDataParallel is not None
    assert gossip.utils.Meter is not None
