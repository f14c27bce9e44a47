"""Multi-process gossiper tests over gloo (CPU).

The push-sum invariant: at every iteration the global sums
``sum_i x_i`` and ``sum_i w_i`` are conserved, and the de-biased
estimates ``x_i / w_i`` converge to the true global average.
(reference behavior: gossip/gossiper.py PushSum/PushPull; usable
standalone per reference README.md:67-68)
"""

import pytest
import torch
import torch.distributed as dist

from tests.dist_utils import run_dist

N = 11  # message length


def _make(rank, world_size, cls_name, graph_cls_name, ppi=1):
    from stochastic_gradient_push_amd import gossiper as G
    from stochastic_gradient_push_amd import graphs

    graph = getattr(graphs, graph_cls_name)(
        rank, world_size, peers_per_itr=ppi
    )
    msg = torch.zeros(N)
    return getattr(G, cls_name)(
        msg, graph=graph, device=torch.device("cpu"),
        rank=rank, world_size=world_size,
    )


def _pushsum_average(rank, world_size, cls_name, graph_cls_name):
    torch.manual_seed(rank)
    gossiper = _make(rank, world_size, cls_name, graph_cls_name)
    x = torch.randn(N)
    w = torch.ones(1)

    # true average via all-reduce
    target = x.clone()
    dist.all_reduce(target)
    target /= world_size

    # non-residual standalone averaging mode: each mix returns the fully
    # mixed (not residual) message
    for _ in range(60):
        x, w = gossiper.mix(x.clone(), w, residual=False)
        x = x.clone()
        w = w.clone()

    est = x / w
    assert torch.allclose(est, target, atol=1e-4), (
        f"rank {rank}: {est} vs {target}"
    )


@pytest.mark.parametrize("graph_cls", [
    "NPeerDynamicDirectedExponentialGraph",
    "DynamicDirectedExponentialGraph",
    "RingGraph",
])
def test_pushsum_converges_to_average(graph_cls):
    run_dist(_pushsum_average, world_size=4,
             args=("PushSum", graph_cls))


def test_pushpull_converges_to_average():
    run_dist(_pushsum_average, world_size=4,
             args=("PushPull", "RingGraph"))


def _mass_conservation(rank, world_size):
    """Residual-mode push-sum: x_i + sum(residuals) keeps global mass."""
    torch.manual_seed(100 + rank)
    gossiper = _make(
        rank, world_size, "PushSum", "NPeerDynamicDirectedExponentialGraph"
    )
    x = torch.randn(N)
    ps_weight = torch.ones(1)

    total0 = x.clone()
    dist.all_reduce(total0)
    w_total0 = torch.tensor([float(world_size)])

    lo = gossiper.mixing_weights["lo"]
    for _ in range(10):
        # residual protocol (as the training wrapper drives it,
        # reference distributed.py:389-434 + 336-387, lazy mixing):
        out = x.clone()
        in_msg, w_recv = gossiper.mix(out, ps_weight, residual=True)
        ps_weight = (ps_weight + w_recv) * lo
        x = (x + in_msg) * lo

        # invariant: global sums conserved
        xs = x.clone()
        dist.all_reduce(xs)
        ws = ps_weight.clone()
        dist.all_reduce(ws)
        assert torch.allclose(xs, total0, atol=1e-4)
        assert torch.allclose(ws, w_total0, atol=1e-5)

    # and the de-biased estimate approaches the average
    est = x / ps_weight
    assert torch.allclose(est, total0 / world_size, atol=5e-2)


def test_pushsum_residual_mass_conserved():
    run_dist(_mass_conservation, world_size=4)


def _bilat(rank, world_size):
    """ws=2: rank 0 passive (persistent async recv), rank 1 active
    (blocking exchange).  The active side does exactly M exchanges; the
    passive side polls until it has completed M, so both exit with no
    outstanding requests."""
    import time

    from stochastic_gradient_push_amd import gossiper as G
    from stochastic_gradient_push_amd.graphs import (
        DynamicBipartiteExponentialGraph,
    )

    torch.manual_seed(rank)
    graph = DynamicBipartiteExponentialGraph(rank, world_size)
    msg = torch.zeros(N)
    gossiper = G.BilatPushPull(
        msg, graph=graph, device=torch.device("cpu"),
        rank=rank, world_size=world_size,
    )
    x = torch.randn(N)
    target = x.clone()
    dist.all_reduce(target)
    target /= world_size

    M = 8
    completions = 0
    deadline = time.time() + 60
    while completions < M:
        assert time.time() < deadline, f"rank {rank} timed out"
        in_msg, completed = gossiper.mix(x.clone())
        if not isinstance(completed, bool) or completed:
            x = (x + in_msg) * 0.5
            completions += 1
        elif gossiper.passive:
            time.sleep(0.005)

    assert torch.allclose(x, target, atol=1e-4), f"rank {rank}: {x} vs {target}"


def test_bilat_pushpull_converges():
    run_dist(_bilat, world_size=2)


@pytest.mark.parametrize("graph_cls", [
    "NPeerDynamicDirectedExponentialGraph",
    "DynamicBipartiteExponentialGraph",
])
def test_pushsum_eight_process_emulation(graph_cls):
    """Multi-node emulation (SURVEY tier): 8 CPU processes, larger
    topology, distributed averaging converges."""
    run_dist(_pushsum_average, world_size=8, args=("PushSum", graph_cls))


def _communicate_helper(rank, world_size):
    """utils.communicate: flatten per dtype, run the op, write back
    (reference helpers.py:73-88)."""
    from stochastic_gradient_push_amd.utils import communicate

    tensors = [
        torch.full((3, 3), float(rank)),
        torch.full((5,), float(rank)).double(),
    ]
    communicate(tensors, lambda tensor: dist.all_reduce(tensor))
    expected = sum(range(world_size))
    for t in tensors:
        assert torch.allclose(t, torch.full_like(t, float(expected)))


def test_communicate_helper():
    run_dist(_communicate_helper, world_size=2)


def _nonregular_wire_weight(rank, world_size):
    """Non-regular graphs transmit the push-sum weight as a trailing
    scalar (reference gossiper.py:83-85, 132): with explicit weight
    tracking the de-biased estimates still converge to the average."""
    from stochastic_gradient_push_amd import gossiper as G
    from stochastic_gradient_push_amd.graphs import (
        NPeerDynamicDirectedExponentialGraph,
    )

    class NonRegularGraph(NPeerDynamicDirectedExponentialGraph):
        def is_regular_graph(self):
            return False  # forces ps-weight on the wire

    torch.manual_seed(rank)
    graph = NonRegularGraph(rank, world_size)
    gossiper = G.PushSum(
        torch.zeros(N), graph=graph, device=torch.device("cpu"),
        rank=rank, world_size=world_size,
    )
    assert not gossiper.regular
    assert gossiper.in_msg_buffer.numel() == N + 1

    x = torch.randn(N)
    w = torch.ones(1)
    target = x.clone()
    dist.all_reduce(target)
    target /= world_size

    for _ in range(60):
        x, w = gossiper.mix(x.clone(), w.clone(), residual=False)
        x = x.clone()
        w = w.clone().reshape(1)

    est = x / w
    assert torch.allclose(est, target, atol=1e-3), (
        f"rank {rank}: max err {(est - target).abs().max()}"
    )


def test_nonregular_wire_weight():
    run_dist(_nonregular_wire_weight, world_size=4)


def _weighted_average(rank, world_size):
    """Non-uniform mixing (WeightedMixing): per-edge weighted messages,
    ps-weight on the wire; de-biased estimates still converge to the
    true average (push-sum is exact for any column-stochastic mixing)."""
    from stochastic_gradient_push_amd import gossiper as G
    from stochastic_gradient_push_amd.graphs import (
        NPeerDynamicDirectedExponentialGraph,
    )
    from stochastic_gradient_push_amd.mixing import WeightedMixing

    torch.manual_seed(20 + rank)
    graph = NPeerDynamicDirectedExponentialGraph(rank, world_size)
    # possible out-peers at distances 2^i; deliberately asymmetric weights
    weights = {
        (rank + 1) % world_size: 0.35,
        (rank + 2) % world_size: 0.20,
    }
    mixing = WeightedMixing(graph, torch.device("cpu"), weights)
    gossiper = G.PushSum(
        torch.zeros(N), graph=graph, mixing=mixing,
        device=torch.device("cpu"), rank=rank, world_size=world_size,
    )
    assert not gossiper.regular  # non-uniform => ps-weight on the wire

    x = torch.randn(N)
    w = torch.ones(1)
    target = x.clone()
    dist.all_reduce(target)
    target /= world_size

    for _ in range(80):
        x, w = gossiper.mix(x.clone(), w.clone(), residual=False)
        x = x.clone()
        w = w.clone().reshape(1)

    est = x / w
    assert torch.allclose(est, target, atol=1e-3), (
        f"rank {rank}: max err {(est - target).abs().max()}"
    )


def test_weighted_mixing_converges():
    run_dist(_weighted_average, world_size=4)


def _weighted_mass_conservation(rank, world_size):
    """Residual protocol with non-uniform mixing: the global sums of x
    and of the push-sum weights are conserved every iteration even
    though lo varies with the rotating peer set."""
    from stochastic_gradient_push_amd import gossiper as G
    from stochastic_gradient_push_amd.graphs import (
        NPeerDynamicDirectedExponentialGraph,
    )
    from stochastic_gradient_push_amd.mixing import WeightedMixing

    torch.manual_seed(40 + rank)
    graph = NPeerDynamicDirectedExponentialGraph(rank, world_size)
    weights = {
        (rank + 1) % world_size: 0.35,
        (rank + 2) % world_size: 0.20,
    }
    mixing = WeightedMixing(graph, torch.device("cpu"), weights)
    gossiper = G.PushSum(
        torch.zeros(N), graph=graph, mixing=mixing,
        device=torch.device("cpu"), rank=rank, world_size=world_size,
    )

    x = torch.randn(N)
    ps = torch.ones(1)
    total0 = x.clone()
    dist.all_reduce(total0)
    w_total0 = torch.tensor([float(world_size)])

    for _ in range(12):
        # explicit (non-lazy) residual protocol: pre-scale by THIS
        # round's lo (exported by the gossiper for the current peer
        # set), exchange, then merge the received residuals
        lo = gossiper.mixing_weights["lo"].clone()
        x = x * lo
        ps = ps * lo
        in_msg, w_recv = gossiper.mix(x.clone(), ps.clone(), residual=True)
        x = x + in_msg
        ps = (ps + w_recv).reshape(1)

        xs = x.clone()
        dist.all_reduce(xs)
        ws = ps.clone()
        dist.all_reduce(ws)
        assert torch.allclose(xs, total0, atol=1e-4), f"rank {rank}"
        assert torch.allclose(ws, w_total0, atol=1e-5), f"rank {rank}"

    est = x / ps
    assert torch.allclose(est, total0 / world_size, atol=0.1)


def test_weighted_mixing_mass_conserved():
    run_dist(_weighted_mass_conservation, world_size=4)


def _chunked_average(rank, world_size):
    """Chunked gossip messages (xGMI link spreading at N>1) pair
    correctly and converge identically."""
    from stochastic_gradient_push_amd import gossiper as G
    from stochastic_gradient_push_amd.graphs import (
        NPeerDynamicDirectedExponentialGraph,
    )

    torch.manual_seed(60 + rank)
    graph = NPeerDynamicDirectedExponentialGraph(rank, world_size)
    gossiper = G.PushSum(
        torch.zeros(N), graph=graph, device=torch.device("cpu"),
        rank=rank, world_size=world_size, chunks=3,
    )
    x = torch.randn(N)
    w = torch.ones(1)
    target = x.clone()
    dist.all_reduce(target)
    target /= world_size
    for _ in range(60):
        x, w = gossiper.mix(x.clone(), w.clone(), residual=False)
        x = x.clone()
        w = w.clone().reshape(1)
    est = x / w
    assert torch.allclose(est, target, atol=1e-4), f"rank {rank}"


def test_chunked_gossip_converges():
    run_dist(_chunked_average, world_size=4)
