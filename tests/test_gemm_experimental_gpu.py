"""wgrad TN kernel numerics (hardware-validated in round 2; promoted
from the gpu_experimental tier to the standard gpu tier)."""

import pytest
import torch

from stochastic_gradient_push_amd import ops

pytestmark = pytest.mark.gpu


def dev():
    return torch.device("cuda", 0)


@pytest.mark.parametrize("shape", [(1024, 128, 64), (25088, 512, 256),
                                   (4096, 96, 72)])
@pytest.mark.parametrize("split", [1, 4])
def test_wgrad_tn_matches_matmul(shape, split):
    M, Co, Ci = shape
    ext = ops._ext_for(torch.empty(1, device=dev()))
    torch.manual_seed(0)
    dy = torch.randn(M, Co, device=dev()).to(torch.bfloat16)
    x = torch.randn(M, Ci, device=dev()).to(torch.bfloat16)
    partials = torch.zeros(split * Co * Ci, device=dev())
    dw = torch.zeros(Co * Ci, device=dev())
    ext.gemm_tn_wgrad_bf16(dy, x, partials, dw, split)
    torch.cuda.synchronize()
    ref = (dy.float().t() @ x.float()).reshape(-1)
    scale = ref.abs().mean() + 1e-3
    err = (dw - ref).abs()
    assert (err.mean() / scale) < 5e-2, (err.mean() / scale).item()
