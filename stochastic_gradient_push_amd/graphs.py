"""Peer-to-peer gossip topologies.

Behavioral parity with the reference graph manager
(reference: gossip/graph_manager.py:35-279) with one architectural change:
edges are plain (src, dest) descriptors instead of owners of dedicated
2-rank ``torch.distributed`` process groups.  The reference pre-created a
broadcast group per directed edge (graph_manager.py:22-32) because NCCL of
that era had no send/recv; on ROCm, RCCL supports true point-to-point
(``ncclSend``/``ncclRecv``), so the gossip layer exchanges messages with
grouped p2p on a single communicator and the graph layer is pure topology
math.  This removes the O(world_size * degree) eager communicator build
(and its 2 warm-up all-reduces per edge) from startup entirely.

Topology semantics (peer sets, rotation order, regular/bipartite/passive/
dynamic predicates) follow reference graph_manager.py:149-279 exactly.
"""

import math
from typing import List, Optional, Tuple


class Edge:
    """A directed gossip edge between two process ranks.

    Parity: reference graph_manager.py:22-32 ``Edge`` minus the per-edge
    process group (see module docstring).  ``src``/``dest`` are *process*
    ranks (node rank x nprocs_per_node), as in the reference.
    """

    __slots__ = ("src", "dest")

    def __init__(self, src: int, dest: int) -> None:
        self.src = src
        self.dest = dest

    def __repr__(self) -> str:
        return f"Edge({self.src}->{self.dest})"

    def __eq__(self, other) -> bool:
        return (
            isinstance(other, Edge)
            and self.src == other.src
            and self.dest == other.dest
        )

    def __hash__(self) -> int:
        return hash((self.src, self.dest))


class GraphManager:
    """Base class for gossip topologies (reference graph_manager.py:35-146).

    ``phone_book[r]`` holds, for every node rank ``r``, the ordered list of
    out-edges ``r`` may send on.  ``peers_per_itr`` out-edges are active per
    iteration; dynamic graphs rotate the active window each iteration.
    """

    def __init__(
        self,
        rank: int,
        world_size: int,
        nprocs_per_node: int = 1,
        local_rank: int = 0,
        peers_per_itr: int = 1,
    ) -> None:
        assert int(peers_per_itr) >= 1
        self.rank = rank
        self.world_size = world_size
        self.nprocs_per_node = nprocs_per_node
        self.local_rank = local_rank
        self.phone_book: List[List[Edge]] = [[] for _ in range(world_size)]
        self._peers_per_itr = peers_per_itr
        self._group_indices = list(range(peers_per_itr))
        self._make_graph()

    # -- construction -----------------------------------------------------

    def _make_graph(self) -> None:
        raise NotImplementedError

    def _add_peers(self, rank: int, peers: List[int]) -> None:
        """Append out-edges rank->peer (dedup, insertion-ordered).

        Parity: reference graph_manager.py:66-73 (ranks scaled by
        nprocs_per_node so src/dest are process ranks).
        """
        book = self.phone_book[rank]
        for peer in peers:
            e = Edge(
                src=rank * self.nprocs_per_node,
                dest=peer * self.nprocs_per_node,
            )
            if e not in book:
                book.append(e)

    # -- predicates --------------------------------------------------------

    def is_regular_graph(self) -> bool:
        raise NotImplementedError

    def is_bipartite_graph(self) -> bool:
        raise NotImplementedError

    def is_passive(self, rank: Optional[int] = None) -> bool:
        raise NotImplementedError

    def is_dynamic_graph(self) -> bool:
        raise NotImplementedError

    # -- peer iteration ----------------------------------------------------

    @property
    def peers_per_itr(self) -> int:
        return self._peers_per_itr

    @peers_per_itr.setter
    def peers_per_itr(self, v: int) -> None:
        self._peers_per_itr = v
        self._group_indices = list(range(v))

    def get_edges(self, rotate: bool = False) -> Tuple[List[Edge], List[Edge]]:
        """Current (out_edges, in_edges) for ``self.rank``.

        Parity: reference graph_manager.py:109-126.  In-edges are found by
        scanning every other rank's phone book at the same group index for
        edges whose dest is this rank.
        """
        if rotate:
            self._rotate_group_indices()
        out_edges: List[Edge] = []
        in_edges: List[Edge] = []
        my_proc_rank = self.rank * self.nprocs_per_node
        for gi in self._group_indices:
            out_edges.append(self.phone_book[self.rank][gi])
            for rank, book in enumerate(self.phone_book):
                if rank == self.rank:
                    continue
                if book[gi].dest == my_proc_rank:
                    in_edges.append(book[gi])
        return out_edges, in_edges

    def get_peers(self, rotate: bool = False) -> Tuple[List[int], List[int]]:
        """Current (out_peers, in_peers) *node* ranks for ``self.rank``.

        Parity: reference graph_manager.py:91-107 (which returned process
        ranks for out-peers and node ranks for in-peers; we return node
        ranks for both — identical when nprocs_per_node == 1, and the
        consistent choice otherwise).
        """
        if rotate:
            self._rotate_group_indices()
        out_peers: List[int] = []
        in_peers: List[int] = []
        my_proc_rank = self.rank * self.nprocs_per_node
        for gi in self._group_indices:
            out_peers.append(
                self.phone_book[self.rank][gi].dest // self.nprocs_per_node
            )
            for rank, book in enumerate(self.phone_book):
                if rank == self.rank:
                    continue
                if book[gi].dest == my_proc_rank:
                    in_peers.append(rank)
        return out_peers, in_peers

    def _rotate_group_indices(self) -> None:
        """Advance the active out-edge window by ``peers_per_itr``
        (reference graph_manager.py:128-133)."""
        inc = self.peers_per_itr
        n = len(self.phone_book[self.rank])
        self._group_indices = [(gi + inc) % n for gi in self._group_indices]

    # -- helpers -----------------------------------------------------------

    def _rotate_forward(self, r: int, p: int) -> int:
        return (r + p) % self.world_size

    def _rotate_backward(self, r: int, p: int) -> int:
        return (r - p) % self.world_size


class DynamicDirectedExponentialGraph(GraphManager):
    """Out-peers at distance +/- 2^i (reference graph_manager.py:149-165)."""

    def _make_graph(self) -> None:
        for rank in range(self.world_size):
            for i in range(int(math.log(self.world_size - 1, 2)) + 1):
                self._add_peers(
                    rank,
                    [
                        self._rotate_forward(rank, 2 ** i),
                        self._rotate_backward(rank, 2 ** i),
                    ],
                )

    def is_regular_graph(self) -> bool:
        return True

    def is_bipartite_graph(self) -> bool:
        return False

    def is_passive(self, rank: Optional[int] = None) -> bool:
        return False

    def is_dynamic_graph(self) -> bool:
        return True


class NPeerDynamicDirectedExponentialGraph(GraphManager):
    """(peers_per_itr+1)-ary exponential out-peer set
    (reference graph_manager.py:167-185).  The default SGP topology."""

    def _make_graph(self) -> None:
        base = self._peers_per_itr + 1
        for rank in range(self.world_size):
            for i in range(int(math.log(self.world_size - 1, base)) + 1):
                for j in range(1, base):
                    dist_to_peer = j * (base ** i)
                    self._add_peers(
                        rank, [self._rotate_forward(rank, dist_to_peer)]
                    )

    def is_regular_graph(self) -> bool:
        return True

    def is_bipartite_graph(self) -> bool:
        return False

    def is_passive(self, rank: Optional[int] = None) -> bool:
        return False

    def is_dynamic_graph(self) -> bool:
        return True


class DynamicBipartiteExponentialGraph(GraphManager):
    """Bipartite exponential graph; even ranks passive
    (reference graph_manager.py:187-216)."""

    def _make_graph(self) -> None:
        for rank in range(self.world_size):
            for i in range(int(math.log(self.world_size - 1, 2)) + 1):
                if i == 0:
                    f_peer = self._rotate_forward(rank, 1)
                    b_peer = self._rotate_backward(rank, 1)
                else:
                    f_peer = self._rotate_forward(rank, 1 + 2 ** i)
                    b_peer = self._rotate_backward(rank, 1 + 2 ** i)
                if not self.is_passive(rank) and (
                    self.is_passive(f_peer) and self.is_passive(b_peer)
                ):
                    self._add_peers(rank, [f_peer, b_peer])
                elif self.is_passive(rank) and not (
                    self.is_passive(f_peer) or self.is_passive(b_peer)
                ):
                    self._add_peers(rank, [f_peer, b_peer])

    def is_regular_graph(self) -> bool:
        return True

    def is_bipartite_graph(self) -> bool:
        return True

    def is_passive(self, rank: Optional[int] = None) -> bool:
        rank = self.rank if rank is None else rank
        return rank % 2 == 0

    def is_dynamic_graph(self) -> bool:
        return True


class DynamicDirectedLinearGraph(GraphManager):
    """Out-peers at every odd distance (reference graph_manager.py:218-236)."""

    def _make_graph(self) -> None:
        for rank in range(self.world_size):
            for i in range(1, self.world_size):
                if i % 2 == 0:
                    continue
                self._add_peers(
                    rank,
                    [
                        self._rotate_forward(rank, i),
                        self._rotate_backward(rank, i),
                    ],
                )

    def is_regular_graph(self) -> bool:
        return True

    def is_bipartite_graph(self) -> bool:
        return False

    def is_passive(self, rank: Optional[int] = None) -> bool:
        return False

    def is_dynamic_graph(self) -> bool:
        return True


class DynamicBipartiteLinearGraph(GraphManager):
    """Bipartite linear graph; even ranks passive
    (reference graph_manager.py:238-263)."""

    def _make_graph(self) -> None:
        for rank in range(self.world_size):
            for i in range(1, self.world_size):
                f_peer = self._rotate_forward(rank, i)
                b_peer = self._rotate_backward(rank, i)
                if not self.is_passive(rank) and (
                    self.is_passive(f_peer) and self.is_passive(b_peer)
                ):
                    self._add_peers(rank, [f_peer, b_peer])
                elif self.is_passive(rank) and not (
                    self.is_passive(f_peer) or self.is_passive(b_peer)
                ):
                    self._add_peers(rank, [f_peer, b_peer])

    def is_regular_graph(self) -> bool:
        return True

    def is_bipartite_graph(self) -> bool:
        return True

    def is_passive(self, rank: Optional[int] = None) -> bool:
        rank = self.rank if rank is None else rank
        return rank % 2 == 0

    def is_dynamic_graph(self) -> bool:
        return True


class RingGraph(GraphManager):
    """Static bidirectional ring (reference graph_manager.py:265-279)."""

    def _make_graph(self) -> None:
        for rank in range(self.world_size):
            self._add_peers(
                rank,
                [
                    self._rotate_forward(rank, 1),
                    self._rotate_backward(rank, 1),
                ],
            )

    def is_regular_graph(self) -> bool:
        return True

    def is_bipartite_graph(self) -> bool:
        return False

    def is_passive(self, rank: Optional[int] = None) -> bool:
        return False

    def is_dynamic_graph(self) -> bool:
        return False
