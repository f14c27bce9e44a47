"""Mixing-weight tests: column stochasticity (mass conservation)."""

import pytest
import torch

from stochastic_gradient_push_amd.graphs import (
    NPeerDynamicDirectedExponentialGraph,
    RingGraph,
)
from stochastic_gradient_push_amd.mixing import UniformMixing


@pytest.mark.parametrize("world_size", [2, 4, 8])
@pytest.mark.parametrize("ppi", [1, 2])
def test_uniform_weights_column_stochastic(world_size, ppi):
    if ppi >= world_size:
        pytest.skip("ppi >= world_size")
    g = NPeerDynamicDirectedExponentialGraph(0, world_size, peers_per_itr=ppi)
    m = UniformMixing(g, torch.device("cpu"))
    w = m.get_mixing_weights(residual_adjusted=False)
    out_peers, _ = g.get_peers()
    total = w["lo"].item() + sum(w[p].item() for p in out_peers)
    assert abs(total - 1.0) < 1e-6


def test_residual_adjusted_weights_are_one():
    g = RingGraph(0, 4)
    m = UniformMixing(g, torch.device("cpu"))
    w = m.get_mixing_weights(residual_adjusted=True)
    assert w["uniform"].item() == 1.0
    assert abs(w["lo"].item() - 0.5) < 1e-6


def test_is_regular():
    g = NPeerDynamicDirectedExponentialGraph(0, 8)
    m = UniformMixing(g, torch.device("cpu"))
    assert m.is_regular() and m.is_uniform()


def test_weighted_mixing_column_stochastic():
    from stochastic_gradient_push_amd.mixing import WeightedMixing

    g = RingGraph(0, 4)
    # possible out-peers of rank 0 in a 4-ring: 1 and 3
    m = WeightedMixing(g, torch.device("cpu"), {1: 0.3, 3: 0.2})
    assert not m.is_uniform() and not m.is_regular()
    for rotate in (False, True, True):
        out_peers, _ = g.get_peers(rotate=rotate)
        w = m.get_mixing_weights(residual_adjusted=False)
        total = w["lo"].item() + sum(w[p].item() for p in out_peers)
        assert abs(total - 1.0) < 1e-6
        # residual-adjusted: w_op/lo with the SAME active-set lo
        wr = m.get_mixing_weights(residual_adjusted=True)
        for p in out_peers:
            assert abs(
                wr[p].item() - w[p].item() / w["lo"].item()
            ) < 1e-6


def test_weighted_mixing_validation():
    from stochastic_gradient_push_amd.mixing import WeightedMixing

    g = RingGraph(0, 4, peers_per_itr=2)  # both ring peers active
    with pytest.raises(ValueError):
        WeightedMixing(g, torch.device("cpu"), {1: -0.1, 3: 0.2})
    m = WeightedMixing(g, torch.device("cpu"), {1: 0.9, 3: 0.3})
    with pytest.raises(ValueError):
        m.get_mixing_weights()  # active weights sum >= 1
    g = RingGraph(0, 4)
    m2 = WeightedMixing(g, torch.device("cpu"), {1: 0.3})
    g.get_peers(rotate=True)  # out-peer becomes 3 -> missing weight
    with pytest.raises(KeyError):
        m2.get_mixing_weights()
