"""Gossip communication primitives (PushSum / PushPull / BilatPushPull).

Behavioral parity with reference gossip/gossiper.py:31-323, rebuilt around
true point-to-point communication:

* The reference emulated p2p with ``dist.broadcast`` on a dedicated 2-rank
  process group per directed edge (reference gossiper.py:194-214,
  graph_manager.py:22-32) — a PyTorch-1.0/NCCL-without-send-recv
  workaround costing O(world_size * degree) eagerly-warmed communicators.
* Here every ``mix()`` issues one ``dist.batch_isend_irecv`` over a single
  communicator, which RCCL executes as a grouped
  ``ncclGroupStart / ncclSend x out / ncclRecv x in / ncclGroupEnd``.  On
  an MI355X node each GPU has 7 direct xGMI links (~153 GB/s each), so a
  node's concurrent per-peer sends+recvs ride *distinct* links in
  parallel — the topology property SGP is designed to exploit.
* Each in-edge receives into its own buffer (pooled, allocated once) so
  receives proceed concurrently; accumulation into ``in_msg_buffer``
  happens after the group completes.

Message layout is unchanged from the reference: one flat 1-D tensor per
dtype, with a trailing push-sum-weight scalar appended only when the
graph/mixing is non-regular (reference gossiper.py:83-85, 132).  The
usable-standalone contract (distributed averaging without NN training,
reference README.md:67-68) is preserved.
"""

import threading
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from .graphs import GraphManager
from .mixing import MixingManager, UniformMixing
from .utils.roctx import trace as _roctx


class _PendingRecv:
    """A posted nonblocking receive with backend-appropriate completion
    detection.

    RCCL/NCCL work objects report ``is_completed()`` accurately (HIP-event
    based).  Gloo work objects only complete inside ``wait()``, so for
    gloo a daemon waiter thread blocks in ``wait()`` and flips an Event.
    """

    def __init__(self, buf: torch.Tensor, src: int, group=None):
        self.req = dist.irecv(tensor=buf, src=src, group=group)
        backend = dist.get_backend(group)
        self._use_thread = backend == "gloo"
        if self._use_thread:
            self._done = threading.Event()
            threading.Thread(target=self._waiter, daemon=True).start()

    def _waiter(self):
        try:
            self.req.wait()
        finally:
            self._done.set()

    def is_completed(self) -> bool:
        if self._use_thread:
            return self._done.is_set()
        return self.req.is_completed()


class Gossiper:
    """Base class for multi-peer gossip averaging."""

    def __init__(
        self,
        msg: torch.Tensor,
        graph: GraphManager,
        device=None,
        mixing: Optional[MixingManager] = None,
        logger=None,
        rank: Optional[int] = None,
        world_size: Optional[int] = None,
        group=None,
        transport=None,
        chunks: int = 1,
    ):
        """
        :param msg: prototype message tensor (sized like the flat params)
        :param graph: GraphManager subclass giving the topology
        :param device: comm device for the message buffers
        :param mixing: MixingManager (default UniformMixing)
        :param logger: optional logger
        :param group: dedicated process group for gossip p2p.  Passing a
            group isolated from the default one lets the gossip thread run
            concurrently with main-thread collectives on a separate RCCL
            communicator (the reference achieved this implicitly via its
            per-edge groups).
        :param transport: optional native
            :class:`~stochastic_gradient_push_amd.comm.RcclTransport`;
            when given, exchanges bypass c10d and go straight through the
            C++ comm core's own RCCL communicator.
        :param chunks: split each peer message into this many chunked
            send/recv pairs inside the same group.  xGMI is point-to-
            point (7 links per MI355X GPU); with peers_per_itr=1 a single
            ~51 MB bf16 message rides however many channels RCCL assigns
            one send — chunking gives the scheduler independent pieces
            to spread across links.  Tune at N>1; 1 = off.
        """
        self.logger = logger
        if rank is None or world_size is None:
            assert dist.is_initialized()
            rank = dist.get_rank()
            world_size = dist.get_world_size()
        self.rank = rank
        self.world_size = world_size
        self.group = group
        self.transport = transport
        assert chunks >= 1
        self.chunks = int(chunks)

        assert isinstance(graph, GraphManager)
        self._graph_manager = graph
        self.passive = self._graph_manager.is_passive()
        self.refresh_peers_(rotate=False)

        self.device = device if device is not None else msg.device
        self.peers_per_itr_device = torch.tensor(
            [self._graph_manager.peers_per_itr],
            device=self.device, dtype=msg.dtype,
        )

        if mixing is None:
            mixing = UniformMixing(self._graph_manager, self.device)
        assert isinstance(mixing, MixingManager)
        self._mixing_manager = mixing
        self.refresh_mixing_weights_()
        self.regular = self._mixing_manager.is_regular()

        # message buffers
        self.out_msg_buffer: List[Tuple[object, torch.Tensor]] = []
        self.in_msg_buffer = msg.clone().detach_().to(self.device)
        self._ps_weight = torch.ones(1, device=self.device, dtype=msg.dtype)
        if not self.regular:
            self.in_msg_buffer = torch.cat(
                [self.in_msg_buffer, self.ps_weight.clone()]
            )
        if self.device.type == "cpu":
            try:
                self.in_msg_buffer = self.in_msg_buffer.pin_memory()
            except Exception as e:  # pin not available everywhere
                if self.logger is not None:
                    self.logger.error(e)
        # one receive buffer per concurrent in-edge (grown on demand);
        # replaces the reference's single serially-reused placeholder
        # (gossiper.py:98) so grouped receives land concurrently.
        self._recv_pool: List[torch.Tensor] = [self.in_msg_buffer.clone()]
        self._send_buffer = self.in_msg_buffer.clone()
        # per-out-edge send buffers, used only by non-uniform mixing
        # (each edge carries a differently-weighted message, reference
        # gossiper.py:125-147 generator semantics)
        self._send_pool_: List[torch.Tensor] = []
        self._pending_req = None

    # -- properties --------------------------------------------------------

    @property
    def ps_weight(self) -> torch.Tensor:
        return self._ps_weight

    @ps_weight.setter
    def ps_weight(self, v) -> None:
        self._ps_weight.data[0] = v

    @property
    def peers_per_itr(self) -> int:
        return self._graph_manager.peers_per_itr

    @peers_per_itr.setter
    def peers_per_itr(self, v: int) -> None:
        self._graph_manager.peers_per_itr = v

    # -- peer / weight refresh ---------------------------------------------

    def refresh_peers_(self, rotate: Optional[bool] = None) -> None:
        if rotate is None:
            rotate = self._graph_manager.is_dynamic_graph()
        assert not (rotate and not self._graph_manager.is_dynamic_graph())
        self.out_edges, self.in_edges = self._graph_manager.get_edges(rotate)

    def refresh_mixing_weights_(self, residual_adjusted: bool = False) -> None:
        self.mixing_weights = self._mixing_manager.get_mixing_weights(
            residual_adjusted
        )

    # -- message prep -------------------------------------------------------

    def _prep_out_msg(
        self, out_msg: torch.Tensor, ps_weight, residual: bool
    ) -> torch.Tensor:
        """Stage the outgoing message (with trailing ps-weight when
        non-regular) into the persistent send buffer and apply the uniform
        mix weight in place.  Counterpart of reference
        ``mix_out_msg_`` (gossiper.py:125-147) minus the generator: with
        true p2p and uniform mixing a single staged buffer serves every
        out-peer in the same RCCL group."""
        self.refresh_mixing_weights_(residual)
        self.ps_weight = ps_weight

        # training fast path (regular + uniform + residual): the
        # residual-adjusted weight is exactly 1 and no ps-weight scalar
        # travels, so the caller's buffer is sent as-is — zero-copy.
        # Safe because the wrapper's flag protocol guarantees the buffer
        # is not mutated until the exchange completes.
        if residual and self.regular and self._mixing_manager.is_uniform():
            return out_msg

        n = out_msg.numel()
        if not self.regular:
            assert self._send_buffer.numel() == n + 1
            self._send_buffer[:n].copy_(out_msg)
            self._send_buffer[n] = self._ps_weight[0].to(out_msg.dtype)
        else:
            assert self._send_buffer.numel() == n
            self._send_buffer.copy_(out_msg)

        if self._mixing_manager.is_uniform():
            if not residual:
                weight = self.mixing_weights["uniform"]
                self._send_buffer.mul_(weight.to(self._send_buffer.dtype))
            # residual-adjusted uniform weight is w/lo == 1.0 exactly (the
            # sender pre-scaled by lo) — no multiply, and no host sync.
            return self._send_buffer

        # non-uniform mixing: one weighted copy per out-edge (the
        # reference's mix_out_msg_ generator, gossiper.py:125-147).
        # _send_buffer holds the unweighted (msg [+ps]) staging.
        sends = self._send_buffers(len(self.out_edges))
        npp = self._graph_manager.nprocs_per_node
        for buf, e in zip(sends, self.out_edges):
            w = self.mixing_weights[e.dest if e.dest in self.mixing_weights
                                    else (e.dest // npp) * npp]
            torch.mul(self._send_buffer, w.to(buf.dtype), out=buf)
        if not residual:
            # loopback share: scale the staging buffer by lo in place so
            # _loopback_msg returns the self-contribution
            self._send_buffer.mul_(
                self.mixing_weights["lo"].to(self._send_buffer.dtype)
            )
        return sends

    def _loopback_msg(self, residual: bool) -> Optional[torch.Tensor]:
        """Self-contribution ``lo * out_msg`` (reference gossiper.py:135-136);
        None in residual mode where the caller keeps its own params.
        With uniform mixing ``lo == w``, so the scaled send buffer IS the
        loopback message — no extra multiply."""
        if residual:
            return None
        return self._send_buffer

    def _recv_buffers(self, n: int) -> List[torch.Tensor]:
        while len(self._recv_pool) < n:
            self._recv_pool.append(self.in_msg_buffer.clone())
        return self._recv_pool[:n]

    def _chunked(self, buf: torch.Tensor) -> List[torch.Tensor]:
        """Split a flat message into ``self.chunks`` contiguous pieces
        (both peers split identically, so chunked sends pair with
        chunked recvs in order within the pairwise channel)."""
        if self.chunks <= 1 or buf.numel() < self.chunks:
            return [buf]
        return list(torch.chunk(buf, self.chunks))

    def _send_buffers(self, n: int) -> List[torch.Tensor]:
        while len(self._send_pool_) < n:
            self._send_pool_.append(self.in_msg_buffer.clone())
        return self._send_pool_[:n]

    # -- cleanup / parse ----------------------------------------------------

    def clean_msg_buffers_(self) -> None:
        """Wait out any in-flight sends (reference gossiper.py:149-158)."""
        while self.out_msg_buffer:
            req, _ = self.out_msg_buffer.pop()
            try:
                req.wait()
            except RuntimeError:
                pass

    def parse_in_msg_buffer(
        self, residual: bool = False
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Split received message and ps-weight (reference
        gossiper.py:160-173).  In the regular case no weight was sent on
        the wire: the receiver infers it as ``own_weight * peers_per_itr``
        (valid because regular mixing keeps all ranks' weights equal)."""
        msg = self.in_msg_buffer
        if not self.regular:
            return msg.narrow(0, 0, msg.numel() - 1), msg[-1]
        if residual:
            return msg, self.ps_weight * self.peers_per_itr_device
        return msg, torch.ones(1, device=self.device, dtype=msg.dtype)

    def mix(self, *args, **kwargs):
        raise NotImplementedError

    # -- the p2p exchange ---------------------------------------------------

    def _exchange(self, send_msg: torch.Tensor, residual: bool) -> None:
        """One grouped p2p round: send ``send_msg`` on every out-edge,
        receive from every in-edge, accumulate into ``in_msg_buffer``.

        All ops go into a single ``batch_isend_irecv`` so RCCL launches
        them inside one ncclGroup and the per-peer transfers proceed
        concurrently on distinct xGMI links.
        """
        loopback = self._loopback_msg(residual)
        if loopback is not None:
            self.in_msg_buffer.copy_(loopback)

        # in-degree-1 residual fast path: receive straight into the
        # accumulator, skipping a zero + add pass over the ~100 MB buffer
        # (reference had the same shortcut, gossiper.py:200-205)
        direct = loopback is None and len(self.in_edges) == 1
        if direct:
            recvs = [self.in_msg_buffer]
        else:
            if loopback is None:
                self.in_msg_buffer.zero_()
            recvs = self._recv_buffers(len(self.in_edges))

        per_edge = isinstance(send_msg, list)
        if self.transport is not None:
            for e in self.out_edges:
                assert e.src == self.rank
            dests = [e.dest for e in self.out_edges]
            srcs = [e.src for e in self.in_edges]
            if per_edge:
                self.transport.exchange_multi(send_msg, dests, recvs, srcs)
            else:
                self.transport.exchange(send_msg, dests, recvs, srcs)
        else:
            ops = []
            for i, e in enumerate(self.out_edges):
                assert e.src == self.rank
                buf = send_msg[i] if per_edge else send_msg
                for piece in self._chunked(buf):
                    ops.append(
                        dist.P2POp(dist.isend, piece, e.dest,
                                   group=self.group)
                    )
            for buf, e in zip(recvs, self.in_edges):
                for piece in self._chunked(buf):
                    ops.append(
                        dist.P2POp(dist.irecv, piece, e.src,
                                   group=self.group)
                    )
            if ops:
                reqs = dist.batch_isend_irecv(ops)
                for r in reqs:
                    r.wait()
        if not direct:
            for buf in recvs:
                self.in_msg_buffer.add_(buf)


class PushSum(Gossiper):
    """Directed push-sum averaging (reference gossiper.py:176-219).

    Column-stochastic mixing: each node pushes weighted shares of its
    (message, weight) pair along out-edges and sums whatever arrives.
    Invariant: the global sums of messages and of push-sum weights are
    conserved every iteration.
    """

    def mix(self, out_msg, ps_weight, residual: bool = False):
        assert out_msg.device.type == self.device.type
        if self.logger is not None:
            self.logger.debug(
                f"in/out -peers {self.in_edges}/{self.out_edges}"
            )
        with _roctx("sgp:gossip_mix"):
            send = self._prep_out_msg(out_msg, ps_weight, residual)
            self._exchange(send, residual)
        self.refresh_peers_()
        # re-derive weights for the NEW peer set so mixing_weights['lo']
        # (read by the wrapper as next round's pre-scale factor) matches
        # the lo the next mix will divide by — load-bearing for
        # non-uniform mixing on dynamic graphs, where lo varies per set
        self.refresh_mixing_weights_(residual)
        self.clean_msg_buffers_()
        return self.parse_in_msg_buffer(residual)


class PushPull(Gossiper):
    """Doubly-stochastic bidirectional averaging (reference
    gossiper.py:222-275).  The reference ordered its blocking
    broadcast-sends/recvs by passive role to avoid deadlock; with grouped
    nonblocking p2p the exchange is symmetric and order-free."""

    def mix(self, out_msg, ps_weight, residual: bool = False):
        assert out_msg.device.type == self.device.type
        if self.logger is not None:
            self.logger.debug(
                f"in/out -peers {self.in_edges}/{self.out_edges}"
            )
        send = self._prep_out_msg(out_msg, ps_weight, residual)
        self._exchange(send, residual)
        self.refresh_peers_()
        # keep mixing_weights['lo'] describing the NEW peer set (same
        # dynamic-graph consistency as PushSum.mix)
        self.refresh_mixing_weights_(residual)
        self.clean_msg_buffers_()
        return self.parse_in_msg_buffer(residual)


class BilatPushPull(Gossiper):
    """Asynchronous bilateral exchange for AD-PSGD (reference
    gossiper.py:278-323).

    Active nodes do a blocking send+recv with their single current peer.
    Passive nodes keep a persistent nonblocking receive posted; when it
    completes they reply to their out-peer.  ``mix`` returns
    ``(in_msg, completed)`` where ``completed=False`` means the passive
    node had nothing to merge this call.
    """

    def mix(self, out_msg: torch.Tensor):
        assert out_msg.device.type == self.device.type
        assert len(self.in_edges) == 1 and len(self.out_edges) == 1
        out_edge, in_edge = self.out_edges[0], self.in_edges[0]
        if self.logger is not None:
            self.logger.debug(f"in/out -edges {in_edge}/{out_edge}")

        if not self.passive:
            send = self._prep_out_msg(out_msg, 1.0, residual=True)
            if isinstance(send, list):
                send = send[0]  # single out-edge by contract
            ops = [
                dist.P2POp(dist.isend, send, out_edge.dest, group=self.group),
                dist.P2POp(
                    dist.irecv, self.in_msg_buffer, in_edge.src,
                    group=self.group,
                ),
            ]
            for r in dist.batch_isend_irecv(ops):
                r.wait()
            completed = True
        else:
            if self._pending_req is None:
                self._pending_req = _PendingRecv(
                    self.in_msg_buffer, in_edge.src, group=self.group
                )
            if self._pending_req.is_completed():
                send = self._prep_out_msg(out_msg, 1.0, residual=True)
                if isinstance(send, list):
                    send = send[0]
                if self.logger is not None:
                    self.logger.debug(f"req. completed; sending to {out_edge}")
                dist.send(tensor=send, dst=out_edge.dest, group=self.group)
                self._pending_req = None
                completed = True
            else:
                completed = False

        if completed:
            self.refresh_peers_()
            self.clean_msg_buffers_()
            return self.parse_in_msg_buffer(residual=True)
        return out_msg, completed
