"""Native comm-core CPU-side behavior (the transport itself needs a GPU;
these check bootstrap + failure modes degrade cleanly)."""

import pytest
import torch

from stochastic_gradient_push_amd import ops


def test_unique_id_blob():
    ext = ops._load_extension()
    if ext is None:
        pytest.skip("extension not built")
    uid = ext.rccl_unique_id()
    assert isinstance(uid, bytes) and len(uid) == 128
    # ids are unique per call
    assert ext.rccl_unique_id() != uid


def test_rccl_comm_without_gpu_raises_cleanly():
    ext = ops._load_extension()
    if ext is None:
        pytest.skip("extension not built")
    if torch.cuda.is_available():
        pytest.skip("GPU present; covered by the gpu-marked test")
    uid = ext.rccl_unique_id()
    with pytest.raises(RuntimeError):
        ext.RcclComm(uid, 0, 1, 0)


def test_bad_unique_id_rejected():
    ext = ops._load_extension()
    if ext is None:
        pytest.skip("extension not built")
    with pytest.raises(RuntimeError):
        ext.RcclComm(b"short", 0, 1, 0)
