"""stochastic_gradient_push_amd — MI355X-native gossip-based distributed SGD.

A from-scratch AMD Instinct MI355X (gfx950/CDNA4) framework with the
capabilities of facebookresearch/stochastic_gradient_push: synchronous and
overlapped Stochastic Gradient Push (SGP/OSGP), decentralized parallel SGD
(D-PSGD), asynchronous decentralized SGD (AD-PSGD) and an all-reduce
baseline, built on PyTorch-ROCm, hand-written HIP/CDNA4 kernels for the
gossip hot path, and RCCL point-to-point over xGMI.

Public API parity with the reference ``gossip`` package
(reference gossip/__init__.py:8-21).
"""

from .ad_psgd import BilatGossipDataParallel
from .distributed import GossipDataParallel
from .gossiper import BilatPushPull, PushPull, PushSum
from .graphs import (
    DynamicBipartiteExponentialGraph,
    DynamicBipartiteLinearGraph,
    DynamicDirectedExponentialGraph,
    DynamicDirectedLinearGraph,
    Edge,
    GraphManager,
    NPeerDynamicDirectedExponentialGraph,
    RingGraph,
)
from .mixing import MixingManager, UniformMixing, WeightedMixing

__version__ = "0.1.0"

__all__ = [
    "BilatGossipDataParallel",
    "GossipDataParallel",
    "BilatPushPull",
    "PushPull",
    "PushSum",
    "Edge",
    "GraphManager",
    "DynamicBipartiteExponentialGraph",
    "DynamicBipartiteLinearGraph",
    "DynamicDirectedExponentialGraph",
    "DynamicDirectedLinearGraph",
    "NPeerDynamicDirectedExponentialGraph",
    "RingGraph",
    "MixingManager",
    "UniformMixing",
    "WeightedMixing",
]
