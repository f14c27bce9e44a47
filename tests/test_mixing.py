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
