"""Native RCCL gossip transport (optional backend).

Default gossip communication goes through ``torch.distributed``
(c10d -> RCCL).  ``RcclTransport`` instead drives RCCL directly from the
C++ comm core (ops/csrc/comm_core.cpp): one dedicated communicator +
HIP stream per transport, one C++ call per gossip round, no c10d
bookkeeping on the hot path.

Select with ``GossipDataParallel(..., comm_backend="rccl")``.
Bootstrap uses the already-initialized torch.distributed world only to
broadcast the 128-byte RCCL unique id.
"""

from typing import List, Optional

import torch
import torch.distributed as dist

from . import ops


class RcclTransport:
    """Thin owner of a native RcclComm (see ops/csrc/comm_core.cpp)."""

    def __init__(self, comm):
        self._comm = comm

    @property
    def rank(self) -> int:
        return self._comm.rank

    def exchange(
        self,
        send: torch.Tensor,
        dests: List[int],
        recvs: List[torch.Tensor],
        srcs: List[int],
        blocking: bool = True,
    ) -> None:
        self._comm.exchange(send, dests, recvs, srcs, blocking)

    def exchange_multi(
        self,
        sends: List[torch.Tensor],
        dests: List[int],
        recvs: List[torch.Tensor],
        srcs: List[int],
        blocking: bool = True,
    ) -> None:
        """Per-destination send buffers (non-uniform mixing)."""
        self._comm.exchange_multi(sends, dests, recvs, srcs, blocking)

    def synchronize(self) -> None:
        self._comm.synchronize()

    def abort(self) -> None:
        self._comm.abort()


def create_rccl_transport(
    device_index: Optional[int] = None,
    rank: Optional[int] = None,
    world_size: Optional[int] = None,
    unique_id: Optional[bytes] = None,
) -> RcclTransport:
    """Collective constructor: every rank of the (torch.distributed)
    world must call this together.  ``unique_id`` may be supplied
    directly to skip the dist-based bootstrap (e.g. single-process
    testing)."""
    ext = ops._load_extension()
    if ext is None:
        raise RuntimeError(
            "native comm core unavailable: HIP extension not built"
        )
    if rank is None or world_size is None:
        assert dist.is_initialized()
        rank = dist.get_rank()
        world_size = dist.get_world_size()
    if device_index is None:
        device_index = torch.cuda.current_device()
    if unique_id is None:
        box = [ext.rccl_unique_id() if rank == 0 else None]
        dist.broadcast_object_list(box, src=0)
        unique_id = box[0]
    return RcclTransport(
        ext.RcclComm(unique_id, rank, world_size, device_index)
    )
