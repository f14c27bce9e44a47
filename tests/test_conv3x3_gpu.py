"""MFMA implicit-GEMM 3x3 conv: numerics vs nn.Conv2d and throughput
vs MIOpen on the ResNet-50 shapes."""

import time

import pytest
import torch
import torch.nn.functional as F

from stochastic_gradient_push_amd import ops

pytestmark = pytest.mark.gpu

CL = torch.channels_last


def dev():
    return torch.device("cuda", 0)


def ext():
    return ops._ext_for(torch.empty(1, device=dev()))


# (N, Ci, H, Co, stride) — every distinct ResNet-50 3x3 shape at bs=8
# plus tails/odd cases
SHAPES = [
    (8, 64, 56, 64, 1),
    (8, 128, 56, 128, 2),
    (8, 128, 28, 128, 1),
    (8, 256, 28, 256, 2),
    (8, 256, 14, 256, 1),
    (8, 512, 14, 512, 2),
    (8, 512, 7, 512, 1),      # M=392 -> tail tile
    (3, 64, 9, 96, 1),        # ragged everything
    (2, 64, 8, 64, 2),
]


@pytest.mark.parametrize("shape", SHAPES)
def test_conv3x3_forward_matches_conv2d(shape):
    n, ci, h, co, stride = shape
    torch.manual_seed(0)
    x = torch.randn(n, ci, h, h, device=dev()).to(torch.bfloat16)
    x = x.contiguous(memory_format=CL)
    w = (torch.randn(co, ci, 3, 3, device=dev()) * 0.1).to(torch.bfloat16)
    ho = (h - 1) // stride + 1
    y = torch.empty(n, co, ho, ho, device=dev(), dtype=torch.bfloat16,
                    memory_format=CL)
    w_pack = w.permute(0, 2, 3, 1).contiguous()
    ext().conv3x3_nhwc_bf16(x, w_pack, y, stride)
    torch.cuda.synchronize()
    ref = F.conv2d(x.float(), w.float(), stride=stride, padding=1)
    err = (y.float() - ref).abs()
    scale = ref.abs().mean() + 1e-3
    assert (err.mean() / scale) < 5e-2, (
        f"{shape}: rel err {(err.mean() / scale).item()}"
    )
    assert torch.allclose(y.float(), ref, atol=2.0, rtol=8e-2), (
        f"{shape}: max err {err.max().item()}"
    )


@pytest.mark.parametrize("stride", [1, 2])
def test_mfma_conv3x3_module_backward(stride):
    from stochastic_gradient_push_amd.models.layers import MfmaConv3x3

    torch.manual_seed(1)
    cin, cout, h = 64, 128, 14
    ref = torch.nn.Conv2d(cin, cout, 3, stride=stride, padding=1,
                          bias=False).to(dev())
    m = MfmaConv3x3(cin, cout, stride=stride).to(dev())
    m.weight.data.copy_(ref.weight.data)

    x = (
        torch.randn(4, cin, h, h, device=dev())
        .to(torch.bfloat16).contiguous(memory_format=CL)
        .requires_grad_(True)
    )
    x2 = x.detach().clone().requires_grad_(True)
    y1 = m(x)
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        y2 = ref(x2)
    assert torch.allclose(y1.float(), y2.float(), atol=8e-2, rtol=8e-2), (
        (y1.float() - y2.float()).abs().max().item()
    )
    dy = torch.randn_like(y1)
    y1.backward(dy)
    y2.backward(dy)
    torch.cuda.synchronize()
    assert torch.allclose(
        x.grad.float(), x2.grad.float(), atol=1e-1, rtol=1e-1
    ), (x.grad.float() - x2.grad.float()).abs().max().item()
    cos = F.cosine_similarity(
        m.weight.grad.flatten().float(),
        ref.weight.grad.flatten().float(), dim=0,
    )
    assert cos.item() > 0.999, cos.item()


def test_conv3x3_throughput_readout():
    """Informational: our kernel vs MIOpen (torch conv) per 3x3 shape
    at bs=32."""

    def t(f, n=20):
        for _ in range(5):
            f()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            f()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    torch.backends.cudnn.benchmark = True
    print()
    for n, ci, h, co, stride in [
        (32, 64, 56, 64, 1),
        (32, 128, 28, 128, 1),
        (32, 256, 14, 256, 1),
        (32, 512, 7, 512, 1),
        (32, 128, 56, 128, 2),
    ]:
        x = torch.randn(n, ci, h, h, device=dev()).to(torch.bfloat16)
        x = x.contiguous(memory_format=CL)
        w = torch.randn(co, ci, 3, 3, device=dev()).to(torch.bfloat16)
        ho = (h - 1) // stride + 1
        y = torch.empty(n, co, ho, ho, device=dev(), dtype=torch.bfloat16,
                        memory_format=CL)
        w_pack = w.permute(0, 2, 3, 1).contiguous()
        fl = 2.0 * n * ho * ho * co * 9 * ci
        t_ours = t(lambda: ext().conv3x3_nhwc_bf16(x, w_pack, y, stride))
        t_lib = t(lambda: F.conv2d(x, w, stride=stride, padding=1))
        print(f"[conv3x3 {n}x{ci}x{h}x{h} -> {co} s{stride}] "
              f"ours {fl / t_ours / 1e12:.0f} TF ({t_ours * 1e6:.0f} us) "
              f"vs miopen {fl / t_lib / 1e12:.0f} TF "
              f"({t_lib * 1e6:.0f} us)")


@pytest.mark.parametrize("shape", [
    (8, 64, 14, 64, 1), (4, 128, 14, 128, 1), (2, 64, 9, 96, 2),
    (8, 64, 7, 128, 1),
])
def test_conv3x3_wgrad_matches_ref(shape):
    """Implicit-TN 3x3 wgrad kernel vs fp32 autograd reference."""
    n, ci, h, co, stride = shape
    torch.manual_seed(2)
    x = torch.randn(n, ci, h, h, device=dev()).to(torch.bfloat16)
    x = x.contiguous(memory_format=CL)
    ho = (h - 1) // stride + 1
    dy = torch.randn(n, co, ho, ho, device=dev()).to(torch.bfloat16)
    dy = dy.contiguous(memory_format=CL)
    dw_flat = torch.empty(co * 9 * ci, device=dev(), dtype=torch.float32)
    ext().conv3x3_wgrad_bf16(x, dy, dw_flat, stride)
    torch.cuda.synchronize()
    dw = dw_flat.view(co, 3, 3, ci).permute(0, 3, 1, 2)

    w = torch.zeros(co, ci, 3, 3, device=dev(), requires_grad=True)
    y = F.conv2d(x.float(), w, stride=stride, padding=1)
    y.backward(dy.float())
    ref = w.grad
    err = (dw - ref).abs()
    scale = ref.abs().mean() + 1e-3
    assert (err.mean() / scale) < 5e-2, (err.mean() / scale).item()
    assert torch.allclose(dw, ref, atol=2.0 * max(1.0, scale.item()),
                          rtol=8e-2), err.max().item()
