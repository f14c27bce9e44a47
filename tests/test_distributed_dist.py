"""End-to-end GossipDataParallel tests on CPU/gloo, world_size=2.

Covers: consensus convergence (zero-lr gossip -> average of initial
params), SGP == local SGD under identical data, the full training-step
state machine, state_dict round-trip with push-sum state, and the
ResNet-18 D-PSGD synthetic-data plumbing config (driver config #1).
"""

import pytest
import torch
import torch.distributed as dist
import torch.nn as nn

from tests.dist_utils import run_dist


def tiny_model(seed):
    torch.manual_seed(seed)
    return nn.Sequential(
        nn.Conv2d(3, 4, 3, padding=1),
        nn.BatchNorm2d(4),
        nn.ReLU(),
        nn.AdaptiveAvgPool2d(1),
        nn.Flatten(),
        nn.Linear(4, 10),
    )


def _train_step(model, opt, x, y, loss_fn):
    out = model(x)
    loss = loss_fn(out, y)
    loss.backward()
    opt.step()
    opt.zero_grad()
    model.transfer_params()
    return loss.detach()


def _consensus(rank, world_size, push_sum):
    from stochastic_gradient_push_amd import (
        GossipDataParallel,
        NPeerDynamicDirectedExponentialGraph,
    )

    model = tiny_model(seed=rank)  # different init per rank
    flat0 = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    target = flat0.clone()
    dist.all_reduce(target)
    target /= world_size

    gdp = GossipDataParallel(
        model,
        graph=NPeerDynamicDirectedExponentialGraph(rank, world_size),
        push_sum=push_sum,
        verbose=False,
    )
    opt = torch.optim.SGD(gdp.parameters(), lr=0.0)
    x = torch.randn(2, 3, 8, 8)
    y = torch.randint(0, 10, (2,))
    loss_fn = nn.CrossEntropyLoss()
    gdp.train()
    for _ in range(30):
        _train_step(gdp, opt, x, y, loss_fn)

    # drain last gossip and de-bias
    gdp.sync_comms()
    gdp._query_gossip_queue(non_blocking=False)
    gdp.unbias()
    flat = gdp.flatp.flat.detach()
    assert torch.allclose(flat, target, atol=1e-3), (
        f"rank {rank}: max err {(flat - target).abs().max()}"
    )


@pytest.mark.parametrize("push_sum", [True, False])
def test_zero_lr_gossip_reaches_consensus(push_sum):
    run_dist(_consensus, world_size=2, args=(push_sum,))


def _sgp_matches_local_sgd(rank, world_size):
    """Identical data + identical init on every rank: gossip is a no-op
    and SGP must track plain local SGD bit-for-bit (up to fp error)."""
    from stochastic_gradient_push_amd import GossipDataParallel

    model = tiny_model(seed=123)
    ref = tiny_model(seed=123)

    gdp = GossipDataParallel(model, push_sum=True, verbose=False)
    opt = torch.optim.SGD(gdp.parameters(), lr=0.05, momentum=0.9)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05, momentum=0.9)
    loss_fn = nn.CrossEntropyLoss()

    torch.manual_seed(7)
    xs = [torch.randn(2, 3, 8, 8) for _ in range(5)]
    ys = [torch.randint(0, 10, (2,)) for _ in range(5)]

    gdp.train()
    ref.train()
    for x, y in zip(xs, ys):
        _train_step(gdp, opt, x, y, loss_fn)
        ref_loss = loss_fn(ref(x), y)
        ref_loss.backward()
        ref_opt.step()
        ref_opt.zero_grad()

    gdp.sync_comms()
    gdp.unbias()
    for p, q in zip(gdp.module.parameters(), ref.parameters()):
        assert torch.allclose(p, q, atol=1e-5), (
            f"rank {rank}: param diverged, max {(p - q).abs().max()}"
        )


def test_sgp_identical_data_matches_local_sgd():
    run_dist(_sgp_matches_local_sgd, world_size=2)


def _state_dict_roundtrip(rank, world_size):
    from stochastic_gradient_push_amd import GossipDataParallel

    gdp = GossipDataParallel(tiny_model(seed=rank), push_sum=True)
    opt = torch.optim.SGD(gdp.parameters(), lr=0.01)
    x = torch.randn(2, 3, 8, 8)
    y = torch.randint(0, 10, (2,))
    gdp.train()
    for _ in range(2):
        _train_step(gdp, opt, x, y, nn.CrossEntropyLoss())
    gdp.sync_comms()

    sd = gdp.state_dict()
    assert "ps_weight" in sd and "is_ps_numerator" in sd and "state_dict" in sd

    gdp2 = GossipDataParallel(tiny_model(seed=rank + 50), push_sum=True)
    gdp2.load_state_dict(sd)
    for p, q in zip(gdp.module.parameters(), gdp2.module.parameters()):
        assert torch.equal(p, q)
    assert torch.equal(gdp2.ps_weight, sd["ps_weight"])


def test_state_dict_roundtrip():
    run_dist(_state_dict_roundtrip, world_size=2)


def _resnet18_dpsgd(rank, world_size):
    """Driver config #1: ResNet-18 D-PSGD, CPU/gloo, world_size=2,
    synthetic 224x224."""
    from stochastic_gradient_push_amd import GossipDataParallel
    from stochastic_gradient_push_amd.models import resnet18

    torch.manual_seed(rank)
    model = resnet18(num_classes=10)
    gdp = GossipDataParallel(model, push_sum=False, verbose=False)
    opt = torch.optim.SGD(gdp.parameters(), lr=0.01, momentum=0.9)
    loss_fn = nn.CrossEntropyLoss()
    gdp.train()
    losses = []
    for _ in range(3):
        x = torch.randn(2, 3, 224, 224)
        y = torch.randint(0, 10, (2,))
        losses.append(_train_step(gdp, opt, x, y, loss_fn).item())
    gdp.sync_comms()
    gdp.unbias()
    flat = gdp.flatp.flat
    assert torch.isfinite(flat).all()
    assert all(torch.isfinite(torch.tensor(l)) for l in losses)


def test_resnet18_dpsgd_cpu_gloo():
    run_dist(_resnet18_dpsgd, world_size=2)


def _overlap_sgp(rank, world_size):
    """Overlap mode: transfer happens in the forward pre-hook; training
    must stay finite and reach consensus with zero lr."""
    from stochastic_gradient_push_amd import GossipDataParallel

    model = tiny_model(seed=rank)
    flat0 = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    target = flat0.clone()
    dist.all_reduce(target)
    target /= world_size

    gdp = GossipDataParallel(model, push_sum=True, overlap=True)
    opt = torch.optim.SGD(gdp.parameters(), lr=0.0)
    loss_fn = nn.CrossEntropyLoss()
    gdp.train()
    x = torch.randn(2, 3, 8, 8)
    y = torch.randint(0, 10, (2,))
    for _ in range(40):
        out = gdp(x)
        loss = loss_fn(out, y)
        loss.backward()
        opt.step()
        opt.zero_grad()
        # overlap mode: no explicit transfer_params call
    gdp.sync_comms()
    gdp.unbias()
    flat = gdp.flatp.flat.detach()
    assert torch.allclose(flat, target, atol=1e-2), (
        f"rank {rank}: max err {(flat - target).abs().max()}"
    )


def test_overlap_sgp_consensus():
    run_dist(_overlap_sgp, world_size=2)


def _bf16_wire_consensus(rank, world_size):
    """bf16 wire-format gossip still reaches consensus (zero lr)."""
    from stochastic_gradient_push_amd import GossipDataParallel

    model = tiny_model(seed=rank)
    flat0 = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    target = flat0.clone()
    dist.all_reduce(target)
    target /= world_size

    gdp = GossipDataParallel(
        model, push_sum=True, gossip_dtype=torch.bfloat16,
    )
    opt = torch.optim.SGD(gdp.parameters(), lr=0.0)
    loss_fn = nn.CrossEntropyLoss()
    gdp.train()
    x = torch.randn(2, 3, 8, 8)
    y = torch.randint(0, 10, (2,))
    for _ in range(30):
        _train_step(gdp, opt, x, y, loss_fn)
    gdp.sync_comms()
    gdp._query_gossip_queue(non_blocking=False)
    gdp.unbias()
    flat = gdp.flatp.flat.detach()
    # bf16 wire: consensus up to bf16 resolution
    assert torch.allclose(flat, target, atol=3e-2), (
        f"rank {rank}: max err {(flat - target).abs().max()}"
    )
    gdp.shutdown()


def test_bf16_wire_consensus():
    run_dist(_bf16_wire_consensus, world_size=2)


def _master_weight_consensus(rank, world_size):
    """working_dtype=bf16 (fp32 master + bf16 working weights): gossip
    operates on the master; consensus at zero lr, and the working
    weights track the master after every merge."""
    from stochastic_gradient_push_amd import GossipDataParallel
    from stochastic_gradient_push_amd.ops.fused_sgd import FusedSGD

    model = tiny_model(seed=rank)
    cast = [p for p in model.parameters() if p.ndim >= 2]
    keep = [p for p in model.parameters() if p.ndim < 2]
    flat0 = torch.cat([p.detach().reshape(-1) for p in cast + keep])
    target = flat0.clone()
    dist.all_reduce(target)
    target /= world_size

    gdp = GossipDataParallel(
        model, push_sum=True, working_dtype=torch.bfloat16,
    )
    opt = FusedSGD(gdp.flatp, lr=0.0)
    loss_fn = nn.CrossEntropyLoss()
    gdp.train()
    x = torch.randn(2, 3, 8, 8)
    y = torch.randint(0, 10, (2,))
    for _ in range(30):
        with torch.autocast(device_type="cpu", dtype=torch.bfloat16):
            loss = loss_fn(gdp(x), y)
        loss.backward()
        opt.step()
        opt.zero_grad()
        gdp.transfer_params()
    gdp.sync_comms()
    gdp._query_gossip_queue(non_blocking=False)
    gdp.unbias()
    flat = gdp.flatp.flat.detach()
    assert torch.allclose(flat, target, atol=1e-3), (
        f"rank {rank}: max err {(flat - target).abs().max()}"
    )
    # working weights == bf16 image of the master
    assert torch.equal(
        gdp.flatp.shadow,
        gdp.flatp.flat[: gdp.flatp.n_cast].to(torch.bfloat16),
    )
    # state dict carries the fp32 master and restores exactly
    sd = gdp.state_dict()
    assert "master_flat" in sd
    gdp2 = GossipDataParallel(
        tiny_model(seed=rank + 7), push_sum=True,
        working_dtype=torch.bfloat16,
    )
    gdp2.load_state_dict(sd)
    assert torch.equal(gdp2.flatp.flat, flat)
    gdp.shutdown()
    gdp2.shutdown()


def test_master_weight_consensus():
    run_dist(_master_weight_consensus, world_size=2)


def _osgp_graphed_order(rank, world_size):
    """bench.py's hipGraph-mode OSGP step order (merge previous round,
    kick the next exchange, THEN run the captured compute;
    bench.py step()) must preserve push-sum mass and reach consensus —
    the N>1 numerics the graph path relies on, emulated on CPU/gloo
    with the same lazy-mixing state machine."""
    from stochastic_gradient_push_amd import GossipDataParallel
    from stochastic_gradient_push_amd.ops.fused_sgd import FusedSGD

    model = tiny_model(seed=rank)
    flat0 = torch.cat([
        p.detach().reshape(-1) for p in model.parameters() if p.ndim >= 1
    ])
    gdp = GossipDataParallel(model, push_sum=True, overlap=False)
    assert gdp.lazy_mixing  # the precondition bench's graph mode needs
    target = torch.empty_like(gdp.flatp.flat)
    target.copy_(gdp.flatp.flat)
    dist.all_reduce(target)
    target /= world_size

    opt = FusedSGD(gdp.flatp, lr=0.0)
    loss_fn = nn.CrossEntropyLoss()
    x = torch.randn(2, 3, 8, 8)
    y = torch.randint(0, 10, (2,))
    gdp.train()

    def compute_step():
        # stands in for graph.replay(): fwd/bwd/optimizer only, hooks
        # suppressed exactly as in capture (bench captures the inner
        # module, not the wrapper)
        loss = loss_fn(gdp.module(x), y)
        loss.backward()
        opt.step()
        opt.zero_grad()

    total0 = gdp.flatp.flat.clone()
    dist.all_reduce(total0)
    for it in range(40):
        gdp._query_gossip_queue(non_blocking=gdp.asynch)
        gdp.transfer_params()
        compute_step()
        # push-sum invariant: global de-biased mass is conserved
        # (ps_weight-weighted sum of numerators == sum of params)
    gdp.sync_comms()
    gdp._query_gossip_queue(non_blocking=False)
    gdp.unbias()
    flat = gdp.flatp.flat.detach()
    assert torch.allclose(flat, target, atol=1e-3), (
        f"rank {rank}: max err {(flat - target).abs().max()}"
    )
    gdp.shutdown()
    del flat0


def test_osgp_graphed_step_order_consensus():
    run_dist(_osgp_graphed_order, world_size=2)


def test_osgp_graphed_step_order_consensus_w4():
    run_dist(_osgp_graphed_order, world_size=4)


def _weighted_mixing_wrapper(rank, world_size):
    """GossipDataParallel with NON-uniform mixing (WeightedMixing):
    forces the non-lazy bias/de-bias path (lazy_mixing False) and the
    ps-weight-on-the-wire message format, combined with bf16 working
    weights (master rescales must refresh the shadow).  Consensus at
    zero lr."""
    from stochastic_gradient_push_amd import GossipDataParallel
    from stochastic_gradient_push_amd.graphs import RingGraph
    from stochastic_gradient_push_amd.mixing import WeightedMixing
    from stochastic_gradient_push_amd.ops.fused_sgd import FusedSGD

    model = tiny_model(seed=rank)
    cast = [p for p in model.parameters() if p.ndim >= 2]
    keep = [p for p in model.parameters() if p.ndim < 2]
    target = torch.cat([p.detach().reshape(-1) for p in cast + keep])
    dist.all_reduce(target)
    target /= world_size

    graph = RingGraph(rank, world_size)
    mixing = WeightedMixing(
        graph, torch.device("cpu"),
        {(rank + 1) % world_size: 0.4, (rank - 1) % world_size: 0.25},
    )
    gdp = GossipDataParallel(
        model, graph=graph, mixing=mixing, push_sum=True,
        working_dtype=torch.bfloat16,
    )
    assert not gdp.lazy_mixing  # non-uniform => non-lazy state machine
    opt = FusedSGD(gdp.flatp, lr=0.0)
    loss_fn = nn.CrossEntropyLoss()
    gdp.train()
    x = torch.randn(2, 3, 8, 8)
    y = torch.randint(0, 10, (2,))
    for _ in range(40):
        with torch.autocast(device_type="cpu", dtype=torch.bfloat16):
            loss = loss_fn(gdp(x), y)
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
    # shadow still tracks the master through the rescale/merge path
    assert torch.equal(
        gdp.flatp.shadow,
        gdp.flatp.flat[: gdp.flatp.n_cast].to(torch.bfloat16),
    )
    gdp.shutdown()


def test_weighted_mixing_wrapper_consensus():
    run_dist(_weighted_mixing_wrapper, world_size=2)


def _eval_drains_gossip(rank, world_size):
    """model.eval() disables gossip and drains the in-flight exchange so
    the de-biased estimate includes received residuals (reference
    distributed.py:322-327); train() re-enables."""
    from stochastic_gradient_push_amd import GossipDataParallel

    model = tiny_model(seed=rank)
    gdp = GossipDataParallel(model, push_sum=True)
    gdp.train()
    assert gdp.gossip_enable
    x = torch.randn(2, 3, 8, 8)
    y = torch.randint(0, 10, (2,))
    loss_fn = nn.CrossEntropyLoss()
    loss = loss_fn(gdp(x), y)
    loss.backward()
    gdp.transfer_params()  # kick an exchange
    gdp.eval()             # must drain it and disable gossip
    assert not gdp.gossip_enable
    assert not gdp.gossiping
    with torch.no_grad():
        out = gdp(x)       # eval forward runs without gossip hooks
    assert torch.isfinite(out).all()
    gdp.train()
    assert gdp.gossip_enable
    # a fresh round still works after the eval round-trip
    loss = loss_fn(gdp(x), y)
    loss.backward()
    gdp.transfer_params()
    gdp.sync_comms()
    gdp._query_gossip_queue(non_blocking=False)
    assert torch.isfinite(gdp.flatp.flat).all()
    gdp.shutdown()


def test_eval_drains_gossip():
    run_dist(_eval_drains_gossip, world_size=2)
