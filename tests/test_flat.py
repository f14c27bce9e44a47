"""FlatParams / FlatBuffer: param re-pointing, grad accumulation into the
flat buffer, optimizer interplay."""

import torch
import torch.nn as nn

from stochastic_gradient_push_amd.ops.flat import FlatBuffer, FlatParams


def small_model():
    torch.manual_seed(0)
    return nn.Sequential(nn.Linear(4, 8), nn.ReLU(), nn.Linear(8, 2))


def test_flat_buffer_views_alias():
    ts = [torch.randn(3, 3), torch.randn(5)]
    orig = [t.clone() for t in ts]
    fb = FlatBuffer(ts)
    for v, o in zip(fb.views, orig):
        assert torch.equal(v, o)
    fb.flat.zero_()
    for v in fb.views:
        assert v.abs().sum() == 0


def test_flat_params_preserve_values_and_training():
    model = small_model()
    ref = small_model()  # same seed -> same weights
    fp = FlatParams(model, flatten_grads=True)

    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.equal(p, q)

    x = torch.randn(16, 4)
    y = torch.randn(16, 2)
    opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    opt_ref = torch.optim.SGD(ref.parameters(), lr=0.1, momentum=0.9)
    for _ in range(5):
        for o, m in ((opt, model), (opt_ref, ref)):
            o.zero_grad()
            loss = ((m(x) - y) ** 2).mean()
            loss.backward()
            o.step()
        if fp.params[0].grad is None or (
            fp.params[0].grad.data_ptr() != fp.flat_grad.data_ptr()
        ):
            fp.rewire_grads()
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p, q, atol=1e-6)


def test_flat_grad_accumulates_into_buffer():
    model = small_model()
    fp = FlatParams(model, flatten_grads=True)
    x = torch.randn(8, 4)
    loss = model(x).sum()
    loss.backward()
    # grads landed in the flat buffer
    assert fp.flat_grad.abs().sum() > 0
    g0 = next(model.parameters()).grad
    assert g0.data_ptr() == fp.flat_grad.data_ptr()


def test_scale_flat_equals_per_param_scale():
    model = small_model()
    ref = small_model()
    fp = FlatParams(model)
    fp.flat.mul_(0.5)
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p, q * 0.5)
