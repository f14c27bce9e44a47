"""Tensor-flattening, communication and logging helpers.

Parity: reference gossip/utils/helpers.py:21-146.  The flatten/unflatten
pair is kept for API compatibility and for callers that hold ad-hoc tensor
lists; the training wrappers themselves use the persistent
:class:`~stochastic_gradient_push_amd.ops.flat.FlatBuffer` instead (one
contiguous device allocation, zero per-step concatenation).
"""

import collections
import logging
import math
import sys
from typing import Dict, Iterable, List, Tuple

import torch
import torch.distributed as dist


def flatten_tensors(tensors: List[torch.Tensor]) -> torch.Tensor:
    """Concatenate dense same-dtype tensors into one 1-D buffer
    (reference helpers.py:21-37)."""
    if len(tensors) == 1:
        return tensors[0].reshape(-1).clone()
    return torch.cat([t.reshape(-1) for t in tensors], dim=0)


def unflatten_tensors(
    flat: torch.Tensor, tensors: Iterable[torch.Tensor]
) -> Tuple[torch.Tensor, ...]:
    """Views into ``flat`` shaped like ``tensors``
    (reference helpers.py:39-57)."""
    out = []
    offset = 0
    for t in tensors:
        n = t.numel()
        out.append(flat.narrow(0, offset, n).view_as(t))
        offset += n
    return tuple(out)


def group_by_dtype(
    tensors: Iterable[torch.Tensor],
) -> Dict[torch.dtype, List[torch.Tensor]]:
    """dtype -> list of tensors (reference helpers.py:60-70)."""
    grouped = collections.defaultdict(list)
    for t in tensors:
        grouped[t.dtype].append(t)
    return grouped


def communicate(tensors: List[torch.Tensor], communication_op) -> None:
    """Flatten per dtype, run ``communication_op(tensor=flat)``, and write
    results back into the original tensors (reference helpers.py:73-88).

    The reference re-pointed each tensor's storage at the flat buffer with
    ``t.set_(f)``; we copy back instead so callers keep ownership of their
    storages (semantically identical, and safe with autograd views).
    """
    for dtype_tensors in group_by_dtype(tensors).values():
        flat = flatten_tensors(dtype_tensors)
        communication_op(tensor=flat)
        for f, t in zip(unflatten_tensors(flat, dtype_tensors), dtype_tensors):
            t.copy_(f)


def make_logger(rank, verbose: bool = True) -> logging.Logger:
    """Per-rank stdout logger, ``rank: LEVEL -- threadName -- msg`` format
    (reference helpers.py:91-114)."""
    logger = logging.getLogger(__name__)
    if not getattr(logger, "handler_set", None):
        console = logging.StreamHandler(stream=sys.stdout)
        console.setFormatter(
            logging.Formatter(
                f"{rank}: %(levelname)s -- %(threadName)s -- %(message)s"
            )
        )
        logger.addHandler(console)
        logger.handler_set = True
    if not getattr(logger, "level_set", None):
        logger.setLevel(logging.DEBUG if verbose else logging.INFO)
        logger.level_set = True
    return logger


def is_power_of(N: int, k: int) -> bool:
    """True when N == k**m for some integer m (reference helpers.py:117-128)."""
    assert isinstance(N, int) and isinstance(k, int)
    assert k >= 0 and N > 0
    if k == 0 and N == 1:
        return True
    if k in (0, 1) and N != 1:
        return False
    return k ** int(round(math.log(N, k))) == N


def create_process_group(ranks: List[int]):
    """Create and warm a new process group (reference helpers.py:131-146).

    The single-element all-reduce forces eager communicator creation so
    later, concurrent first-use does not skew across ranks (RCCL inherits
    NCCL's lazy-init behavior).
    """
    init = torch.ones(1)
    if torch.cuda.is_available():
        init = init.cuda()
    group = dist.new_group(ranks)
    if dist.get_rank() in ranks:
        dist.all_reduce(init, group=group)
    return group
