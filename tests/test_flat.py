"""FlatParams / FlatBuffer: param re-pointing, grad accumulation into the
flat buffer, optimizer interplay."""

import pytest
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


def test_flatparams_working_dtype_layout():
    """bf16 working weights: matrix params re-pointed at the bf16 shadow
    (cast section first in the master), BN-style vectors stay fp32."""
    import torch.nn as nn

    from stochastic_gradient_push_amd.ops.flat import FlatParams

    torch.manual_seed(0)
    m = nn.Sequential(nn.Linear(8, 16), nn.BatchNorm1d(16), nn.Linear(16, 4))
    ref = [p.detach().clone() for p in m.parameters() if p.requires_grad]
    fp = FlatParams(m, flatten_grads=True, working_dtype=torch.bfloat16)

    assert fp.flat.dtype == torch.float32
    assert fp.shadow is not None and fp.shadow.dtype == torch.bfloat16
    n_cast_expect = 8 * 16 + 16 * 4
    assert fp.n_cast == n_cast_expect
    # matrix params are bf16 shadow views; vectors fp32 master views
    for p in m.parameters():
        if p.ndim >= 2:
            assert p.dtype == torch.bfloat16
            assert p.grad.dtype == torch.bfloat16
        else:
            assert p.dtype == torch.float32
            assert p.grad.dtype == torch.float32
    # master holds the original fp32 values (cast-section reordered)
    flat_ref = torch.cat(
        [r.reshape(-1) for r in ref if r.ndim >= 2]
        + [r.reshape(-1) for r in ref if r.ndim < 2]
    )
    assert torch.equal(fp.flat, flat_ref)
    # shadow is the bf16 image of the master's cast section
    assert torch.equal(fp.shadow, fp.flat[:fp.n_cast].to(torch.bfloat16))

    # sync_shadow propagates master edits
    with torch.no_grad():
        fp.flat.mul_(2.0)
    fp.sync_shadow()
    assert torch.equal(fp.shadow, fp.flat[:fp.n_cast].to(torch.bfloat16))

    assert fp.grads_wired()
    for p in m.parameters():
        p.grad = None
    assert not fp.grads_wired()
    fp.rewire_grads()
    assert fp.grads_wired()


def test_fused_sgd_mixed_matches_torch():
    """FusedSGD in master-weight mode == torch.optim.SGD on an fp32
    model fed the same (bf16-rounded) gradients."""
    import torch.nn as nn

    from stochastic_gradient_push_amd.ops.flat import FlatParams
    from stochastic_gradient_push_amd.ops.fused_sgd import FusedSGD

    torch.manual_seed(1)
    m = nn.Sequential(nn.Linear(6, 8), nn.BatchNorm1d(8), nn.Linear(8, 3))
    import copy

    oracle = copy.deepcopy(m)
    fp = FlatParams(m, flatten_grads=True, working_dtype=torch.bfloat16)
    opt = FusedSGD(fp, lr=0.1, momentum=0.9, weight_decay=1e-4)
    oracle_params = (
        [p for p in oracle.parameters() if p.ndim >= 2]
        + [p for p in oracle.parameters() if p.ndim < 2]
    )
    oopt = torch.optim.SGD(
        oracle_params, lr=0.1, momentum=0.9, weight_decay=1e-4
    )

    x = torch.randn(5, 6)
    y = torch.randn(5, 3)
    for step in range(3):
        opt.zero_grad()
        with torch.autocast(device_type="cpu", dtype=torch.bfloat16):
            loss = ((m(x) - y) ** 2).mean()
        loss.backward()
        # oracle gets the SAME grads the mixed optimizer sees
        with torch.no_grad():
            offset = 0
            for p in oracle_params:
                n = p.numel()
                if p.ndim >= 2:
                    p.grad = fp.flat_grad_w.narrow(0, offset, n).view_as(
                        p
                    ).float()
                else:
                    p.grad = fp.flat_grad.narrow(
                        0, offset - fp.n_cast, n
                    ).view_as(p).clone()
                offset += n
        opt.step()
        oopt.step()
        oracle_flat = torch.cat(
            [p.detach().reshape(-1) for p in oracle_params]
        )
        assert torch.allclose(fp.flat, oracle_flat, atol=1e-5), (
            step, (fp.flat - oracle_flat).abs().max()
        )
        # working weights track the master
        assert torch.equal(
            fp.shadow, fp.flat[:fp.n_cast].to(torch.bfloat16)
        )


@pytest.mark.parametrize("working", [None, torch.bfloat16])
def test_steal_mode_matches_wired(working):
    """FusedSGD(steal_grads=True) (assign + one fused gather) must be
    numerically identical to the wired-views accumulate path."""
    import copy

    import torch.nn as nn

    from stochastic_gradient_push_amd.ops.flat import FlatParams
    from stochastic_gradient_push_amd.ops.fused_sgd import FusedSGD

    torch.manual_seed(3)
    m1 = nn.Sequential(nn.Linear(6, 8), nn.BatchNorm1d(8), nn.Linear(8, 3))
    m2 = copy.deepcopy(m1)
    fp1 = FlatParams(m1, flatten_grads=True, working_dtype=working)
    fp2 = FlatParams(m2, flatten_grads=True, working_dtype=working)
    o1 = FusedSGD(fp1, lr=0.1, momentum=0.9, weight_decay=1e-4)
    o2 = FusedSGD(fp2, lr=0.1, momentum=0.9, weight_decay=1e-4,
                  steal_grads=True)

    x = torch.randn(5, 6)
    y = torch.randn(5, 3)
    for _ in range(3):
        for m, o in ((m1, o1), (m2, o2)):
            o.zero_grad()
            if working is not None:
                with torch.autocast(device_type="cpu",
                                    dtype=torch.bfloat16):
                    loss = ((m(x) - y) ** 2).mean()
            else:
                loss = ((m(x) - y) ** 2).mean()
            loss.backward()
            o.step()
        assert torch.allclose(fp1.flat, fp2.flat, atol=1e-6), (
            (fp1.flat - fp2.flat).abs().max()
        )
