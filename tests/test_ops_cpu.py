"""CPU (torch-fallback) numerics for the fused-op layer.

These same semantics are the oracle for the HIP kernels in
test_ops_gpu.py.
"""

import pytest
import torch

from stochastic_gradient_push_amd import ops


def test_scale_():
    x = torch.randn(100)
    ref = x * 0.25
    ops.scale_(x, 0.25)
    assert torch.allclose(x, ref)


def test_scale_tensor_scalar():
    x = torch.randn(100)
    a = torch.tensor([0.5])
    ref = x * 0.5
    ops.scale_(x, a)
    assert torch.allclose(x, ref)


def test_add_scale_():
    x = torch.randn(64)
    r = torch.randn(64)
    ref = (x + r) * 0.5
    ops.add_scale_(x, r, 0.5)
    assert torch.allclose(x, ref)


def test_pack_mix_():
    x = torch.randn(64)
    out = torch.empty_like(x)
    ref = x * 0.3
    ops.pack_mix_(x, out, 0.3)
    assert torch.allclose(x, ref)
    assert torch.allclose(out, ref)


def test_average_():
    x = torch.randn(64)
    y = torch.randn(64)
    ref = (x + y) / 2
    ops.average_(x, y)
    assert torch.allclose(x, ref)


@pytest.mark.parametrize("nesterov", [False, True])
@pytest.mark.parametrize("momentum", [0.0, 0.9])
def test_sgd_step_matches_torch(momentum, nesterov):
    if nesterov and momentum == 0.0:
        pytest.skip("torch requires momentum for nesterov")
    torch.manual_seed(1)
    n = 257
    p = torch.randn(n)
    p_ref = torch.nn.Parameter(p.clone())
    opt = torch.optim.SGD(
        [p_ref], lr=0.1, momentum=momentum, weight_decay=1e-4,
        nesterov=nesterov,
    )
    buf = torch.zeros(n)
    for step in range(4):
        g = torch.randn(n)
        p_ref.grad = g.clone()
        opt.step()
        ops.sgd_step_(
            p, g.clone(), buf, lr=0.1, momentum=momentum,
            weight_decay=1e-4, nesterov=nesterov, first_step=(step == 0),
        )
        assert torch.allclose(p, p_ref.detach(), atol=1e-6), f"step {step}"


def test_pack_mix_cast_and_add_scale_cast():
    x = torch.randn(65)
    out = torch.empty(65, dtype=torch.bfloat16)
    ref = x * 0.5
    ops.pack_mix_cast_(x, out, 0.5)
    assert torch.allclose(x, ref)
    assert torch.allclose(out.float(), ref, atol=1e-2, rtol=1e-2)

    y = torch.randn(65)
    r = torch.randn(65).to(torch.bfloat16)
    ref2 = (y + r.float()) * 0.25
    ops.add_scale_cast_(y, r, 0.25)
    assert torch.allclose(y, ref2, atol=1e-6)
