"""Mixing-weight managers for gossip averaging.

Parity: reference gossip/mixing_manager.py:19-56.  Weights form a
column-stochastic mixing matrix: each node keeps ``lo`` of its own mass and
sends weight-``w`` shares to each active out-peer, with
``lo + sum_op(w_op * lo) == 1`` in the residual-adjusted form.
"""

from typing import Dict

import torch

from .graphs import GraphManager


class MixingManager:
    def __init__(self, graph: GraphManager, device) -> None:
        self.graph_manager = graph
        self.device = device

    def is_regular(self) -> bool:
        """True when no bias accumulates in the stationary distribution of
        the mixing matrix, i.e. push-sum weights need not be tracked
        explicitly (reference mixing_manager.py:25-31)."""
        return self.graph_manager.is_regular_graph() and self.is_uniform()

    def is_uniform(self) -> bool:
        raise NotImplementedError

    def get_mixing_weights(
        self, residual_adjusted: bool = True
    ) -> Dict[object, torch.Tensor]:
        raise NotImplementedError


class UniformMixing(MixingManager):
    """Uniform 1/(out_degree + 1) mixing (reference mixing_manager.py:41-56).

    Returns a dict with keys ``'lo'`` (self weight), ``'uniform'`` (the
    per-out-peer weight) and one entry per out-peer *process* rank.  When
    ``residual_adjusted`` the out-peer weights are divided by ``lo`` because
    the sender pre-scales its whole message by ``lo`` before transmitting
    (the residual form used by the training wrapper).
    """

    def get_mixing_weights(
        self, residual_adjusted: bool = True
    ) -> Dict[object, torch.Tensor]:
        out_peers, _ = self.graph_manager.get_peers()
        n = len(out_peers) + 1.0
        w = torch.tensor([1.0 / n], device=self.device)
        weights: Dict[object, torch.Tensor] = {"lo": w.clone()}
        w_op = (w / weights["lo"]) if residual_adjusted else w
        weights["uniform"] = w_op.clone()
        npp = self.graph_manager.nprocs_per_node
        for op in out_peers:
            weights[op * npp] = w_op.clone()
        return weights

    def is_uniform(self) -> bool:
        return True


class WeightedMixing(MixingManager):
    """Non-uniform column-stochastic mixing with explicit per-peer weights.

    The reference's ``MixingManager`` is an extension point whose message
    machinery (``mix_out_msg_``, reference gossiper.py:125-147) supports
    per-edge weights even though only ``UniformMixing`` ships; this class
    restores that capability here.

    :param weights: dict mapping each possible out-peer NODE rank to its
        mixing weight.  Column-stochasticity must hold per iteration over
        the graph's *currently active* out-peers: the self-weight is
        ``lo = 1 - sum(weights[p] for active p)`` and must stay positive
        for every peer set the graph can activate.

    Non-uniform mixing is never "regular", so the push-sum weight always
    travels on the wire (scaled per edge, like the message itself).
    """

    def __init__(self, graph: GraphManager, device, weights: Dict[int, float]):
        super().__init__(graph, device)
        if any(w <= 0 for w in weights.values()):
            raise ValueError("all mixing weights must be positive")
        self._peer_weights = {int(k): float(v) for k, v in weights.items()}

    def is_uniform(self) -> bool:
        return False

    def get_mixing_weights(
        self, residual_adjusted: bool = True
    ) -> Dict[object, torch.Tensor]:
        out_peers, _ = self.graph_manager.get_peers()
        missing = [op for op in out_peers if op not in self._peer_weights]
        if missing:
            raise KeyError(
                f"no mixing weight for current out-peer(s) {missing}; "
                "WeightedMixing needs a weight for every peer the graph "
                "can activate"
            )
        total = sum(self._peer_weights[op] for op in out_peers)
        if not total < 1.0 - 1e-9:
            raise ValueError(
                f"active out-peer weights sum to {total}; must be < 1 so "
                "the self-weight stays positive"
            )
        lo = torch.tensor([1.0 - total], device=self.device)
        weights: Dict[object, torch.Tensor] = {"lo": lo.clone()}
        npp = self.graph_manager.nprocs_per_node
        for op in out_peers:
            w = torch.tensor([self._peer_weights[op]], device=self.device)
            weights[op * npp] = (w / lo) if residual_adjusted else w
        return weights
