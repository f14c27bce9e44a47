"""Drop-in import alias: the reference package was named ``gossip``
(reference gossip/__init__.py:8-21).  All implementations live in
``stochastic_gradient_push_amd``."""

from stochastic_gradient_push_amd import (
    BilatGossipDataParallel,
    BilatPushPull,
    DynamicBipartiteExponentialGraph,
    DynamicBipartiteLinearGraph,
    DynamicDirectedExponentialGraph,
    DynamicDirectedLinearGraph,
    Edge,
    GossipDataParallel,
    GraphManager,
    MixingManager,
    NPeerDynamicDirectedExponentialGraph,
    PushPull,
    PushSum,
    RingGraph,
    UniformMixing,
)

__all__ = [
    "BilatGossipDataParallel", "GossipDataParallel", "BilatPushPull",
    "PushPull", "PushSum", "Edge", "GraphManager",
    "DynamicBipartiteExponentialGraph", "DynamicBipartiteLinearGraph",
    "DynamicDirectedExponentialGraph", "DynamicDirectedLinearGraph",
    "NPeerDynamicDirectedExponentialGraph", "RingGraph",
    "MixingManager", "UniformMixing",
]
