"""rocTX range markers around the gossip/fused-kernel hot path.

On ROCm, ``torch.cuda.nvtx`` is backed by rocTX (roctracer), so these
ranges appear in ``rocprofv3`` marker traces alongside the kernel rows
(the reference had no tracing beyond wall-clock Meters;
reference gossip_sgd.py:256-274).

Disable with SGP_ROCTX=0 (ranges cost ~half a microsecond each on the
host; they are never recorded inside hipGraph capture anyway).
"""

import contextlib
import os

import torch

_ENABLED = (
    os.environ.get("SGP_ROCTX", "1") != "0" and torch.cuda.is_available()
)


@contextlib.contextmanager
def trace(name: str):
    if _ENABLED:
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


def mark(name: str) -> None:
    if _ENABLED:
        torch.cuda.nvtx.mark(name)
