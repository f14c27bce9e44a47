"""Topology unit tests (pure functions of rank/world_size).

Checks peer sets, rotation, degree-regularity and the bipartite
passive/active assignment for all six graph families
(reference semantics: gossip/graph_manager.py:149-279).
"""

import pytest

from stochastic_gradient_push_amd.graphs import (
    DynamicBipartiteExponentialGraph,
    DynamicBipartiteLinearGraph,
    DynamicDirectedExponentialGraph,
    DynamicDirectedLinearGraph,
    NPeerDynamicDirectedExponentialGraph,
    RingGraph,
)

ALL_GRAPHS = [
    DynamicDirectedExponentialGraph,
    NPeerDynamicDirectedExponentialGraph,
    DynamicBipartiteExponentialGraph,
    DynamicDirectedLinearGraph,
    DynamicBipartiteLinearGraph,
    RingGraph,
]


@pytest.mark.parametrize("cls", ALL_GRAPHS)
@pytest.mark.parametrize("world_size", [2, 4, 8])
def test_out_in_degree_balance(cls, world_size):
    """At every group index, total out-degree == total in-degree and each
    rank has exactly peers_per_itr out-edges."""
    graphs = [cls(r, world_size) for r in range(world_size)]
    for g in graphs:
        out_edges, in_edges = g.get_edges(rotate=False)
        assert len(out_edges) == g.peers_per_itr
        for e in out_edges:
            assert e.src == g.rank
            assert 0 <= e.dest < world_size
            assert e.dest != g.rank
        # regular graphs: in-degree == out-degree at every step
        if g.is_regular_graph():
            assert len(in_edges) == len(out_edges)


@pytest.mark.parametrize("cls", ALL_GRAPHS)
@pytest.mark.parametrize("world_size", [4, 8])
def test_edges_globally_consistent(cls, world_size):
    """Every send has exactly one matching receive: union over ranks of
    out-edges equals union of in-edges at each rotation step."""
    graphs = [cls(r, world_size) for r in range(world_size)]
    steps = 5 if graphs[0].is_dynamic_graph() else 1
    for step in range(steps):
        rotate = step > 0
        outs, ins = set(), set()
        for g in graphs:
            o, i = g.get_edges(rotate=rotate)
            outs.update((e.src, e.dest) for e in o)
            ins.update((e.src, e.dest) for e in i)
        assert outs == ins


@pytest.mark.parametrize("world_size", [4, 8, 16])
def test_rotation_cycles_through_phone_book(world_size):
    g = NPeerDynamicDirectedExponentialGraph(0, world_size)
    seen = set()
    booklen = len(g.phone_book[0])
    for _ in range(booklen):
        out, _ = g.get_edges(rotate=True)
        seen.add(out[0].dest)
    # after booklen rotations every out-peer has been used
    assert len(seen) == booklen


def test_exponential_peers_powers_of_two():
    g = DynamicDirectedExponentialGraph(0, 8)
    dests = {e.dest for e in g.phone_book[0]}
    assert dests == {1, 2, 4, 7, 6}  # +-1, +-2, +-4 mod 8 (4 == -4)


def test_npeer_default_peers():
    # world_size 8, 1 peer/itr: out-peers at distance 2^i
    g = NPeerDynamicDirectedExponentialGraph(0, 8)
    dests = [e.dest for e in g.phone_book[0]]
    assert dests == [1, 2, 4]


def test_npeer_two_peers_per_itr():
    g = NPeerDynamicDirectedExponentialGraph(0, 9, peers_per_itr=2)
    dests = [e.dest for e in g.phone_book[0]]
    # base 3: j*(3^i) for i in 0..1, j in 1..2 -> 1,2,3,6
    assert dests == [1, 2, 3, 6]
    out, _ = g.get_edges(rotate=False)
    assert [e.dest for e in out] == [1, 2]


@pytest.mark.parametrize(
    "cls", [DynamicBipartiteExponentialGraph, DynamicBipartiteLinearGraph]
)
def test_bipartite_edges_cross_parity(cls):
    world_size = 8
    graphs = [cls(r, world_size) for r in range(world_size)]
    for g in graphs:
        assert g.is_bipartite_graph()
        assert g.is_passive() == (g.rank % 2 == 0)
        for e in g.phone_book[g.rank]:
            assert (e.src % 2) != (e.dest % 2)


def test_ring_static():
    g = RingGraph(2, 8)
    assert not g.is_dynamic_graph()
    out, inn = g.get_edges(rotate=False)
    assert {e.dest for e in out} == {3}
    assert {e.src for e in inn} == {1}


def test_peers_match_edges():
    g = NPeerDynamicDirectedExponentialGraph(3, 8)
    out_p, in_p = g.get_peers()
    out_e, in_e = g.get_edges()
    assert out_p == [e.dest for e in out_e]
    assert in_p == [e.src for e in in_e]
