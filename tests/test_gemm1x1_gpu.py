"""Hand-written MFMA NT GEMM (1x1-conv shape): layout probe, numerics vs
torch.matmul, and a quick throughput readout."""

import time

import pytest
import torch

from stochastic_gradient_push_amd import ops

pytestmark = pytest.mark.gpu


def dev():
    return torch.device("cuda", 0)


def ext():
    return ops._ext_for(torch.empty(1, device=dev()))


def test_mfma_fragment_layout_probe():
    """Empirically verify the assumed 16x16x32 bf16 fragment mapping:
    one MFMA from global memory must reproduce A @ B^T exactly."""
    torch.manual_seed(0)
    A = torch.randn(16, 32, device=dev()).to(torch.bfloat16)
    # asymmetric B so a transposed C-write cannot pass (guide rule)
    B = (torch.randn(16, 32, device=dev()) * torch.linspace(
        0.1, 2.0, 32, device=dev()
    )).to(torch.bfloat16)
    C = torch.zeros(16, 16, device=dev())
    ext().mfma_probe(A.contiguous(), B.contiguous(), C)
    torch.cuda.synchronize()
    ref = A.float() @ B.float().t()
    assert torch.allclose(C, ref, atol=2e-1, rtol=2e-2), (
        f"layout mismatch, max err {(C - ref).abs().max()}"
    )


RESNET_SHAPES = [
    # (M, N, K): 1x1-conv GEMMs of ResNet-50 at bs=32
    (100352, 64, 256),
    (100352, 256, 64),
    (25088, 128, 512),
    (25088, 512, 128),
    (6272, 1024, 256),
    (1568, 2048, 512),
    (1000, 100, 32),      # ragged M/N
    (130, 72, 64),        # tails everywhere
]


@pytest.mark.parametrize("shape", RESNET_SHAPES)
def test_gemm_nt_matches_matmul(shape):
    M, N, K = shape
    torch.manual_seed(1)
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    ext().gemm_nt_bf16(A, B, C)
    torch.cuda.synchronize()
    ref = (A.float() @ B.float().t()).to(torch.bfloat16)
    err = (C.float() - ref.float()).abs()
    scale = ref.float().abs().mean() + 1e-3
    assert (err.mean() / scale) < 5e-2, (
        f"{shape}: rel mean err {(err.mean() / scale).item()}"
    )
    # spot-exactness vs fp32 reference within bf16 tolerance
    assert torch.allclose(
        C.float(), ref.float(), atol=2.0, rtol=8e-2
    ), f"{shape}: max err {err.max()}"


def test_gemm_nt_throughput_readout():
    """Informational: TFLOP/s of the hand-written kernel vs torch.matmul
    on a mid-size 1x1 shape (printed to the pytest log)."""
    M, N, K = 25088, 512, 512
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)

    def t(f, n=30):
        for _ in range(5):
            f()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            f()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    flops = 2.0 * M * N * K
    t_ours = t(lambda: ext().gemm_nt_bf16(A, B, C))
    Bt = B.t().contiguous().t()  # matmul-friendly layout
    t_lib = t(lambda: torch.matmul(A, Bt))
    print(
        f"\n[gemm {M}x{N}x{K}] ours {flops / t_ours / 1e12:.1f} TF "
        f"({t_ours * 1e6:.0f} us) vs torch.matmul "
        f"{flops / t_lib / 1e12:.1f} TF ({t_lib * 1e6:.0f} us)"
    )
    assert t_ours < 1.0  # sanity only


@pytest.mark.parametrize("shape", [(25088, 512, 512), (1568, 2048, 512),
                                   (130, 72, 64)])
def test_gemm_nt_v2_matches_v1(shape):
    M, N, K = shape
    torch.manual_seed(2)
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C1 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    C2 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    ext().gemm_nt_bf16(A, B, C1)
    ext().gemm_nt_bf16_v2(A, B, C2)
    torch.cuda.synchronize()
    assert torch.equal(C1, C2)


def test_gemm_v2_throughput_readout():
    M, N, K = 25088, 512, 512
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)

    def t(f, n=30):
        for _ in range(5):
            f()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            f()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    flops = 2.0 * M * N * K
    t1 = t(lambda: ext().gemm_nt_bf16(A, B, C))
    t2 = t(lambda: ext().gemm_nt_bf16_v2(A, B, C))
    print(f"\n[gemm v1 vs v2] {flops / t1 / 1e12:.1f} TF vs "
          f"{flops / t2 / 1e12:.1f} TF")


@pytest.mark.parametrize("shape", [(25088, 512, 512), (1568, 2048, 512),
                                   (256, 64, 64), (130, 72, 64)])
def test_gemm_nt_v3_matches_v1(shape):
    M, N, K = shape
    torch.manual_seed(3)
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C1 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    C3 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    ext().gemm_nt_bf16(A, B, C1)
    ext().gemm_nt_bf16_v3(A, B, C3)
    torch.cuda.synchronize()
    assert torch.equal(C1, C3), (
        (C1.float() - C3.float()).abs().max().item()
    )


def test_gemm_v3_throughput_readout():
    M, N, K = 25088, 512, 512
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)

    def t(f, n=30):
        for _ in range(5):
            f()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            f()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    flops = 2.0 * M * N * K
    t2 = t(lambda: ext().gemm_nt_bf16_v2(A, B, C))
    t3 = t(lambda: ext().gemm_nt_bf16_v3(A, B, C))
    print(f"\n[gemm v2 vs v3-glds] {flops / t2 / 1e12:.1f} TF vs "
          f"{flops / t3 / 1e12:.1f} TF")


@pytest.mark.parametrize("shape", [(25088, 512, 512), (1568, 2048, 512),
                                   (256, 64, 64), (130, 72, 64)])
def test_gemm_nt_v5_matches_v1(shape):
    """v5 = v3 + XCD-aware tile remap: numerics must be bitwise v1."""
    M, N, K = shape
    torch.manual_seed(5)
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C1 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    C5 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    ext().gemm_nt_bf16(A, B, C1)
    ext().gemm_nt_bf16_v5(A, B, C5)
    torch.cuda.synchronize()
    assert torch.equal(C1, C5), (
        (C1.float() - C5.float()).abs().max().item()
    )


def test_gemm_v5_throughput_readout():
    M, N, K = 25088, 512, 512
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)

    def t(f, n=30):
        for _ in range(5):
            f()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            f()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    flops = 2.0 * M * N * K
    t3 = t(lambda: ext().gemm_nt_bf16_v3(A, B, C))
    t5 = t(lambda: ext().gemm_nt_bf16_v5(A, B, C))
    print(f"\n[gemm v3 vs v5-xcd] {flops / t3 / 1e12:.1f} TF vs "
          f"{flops / t5 / 1e12:.1f} TF")


@pytest.mark.parametrize("span", [False, True])
@pytest.mark.parametrize("shape", [(25088, 512, 512), (12800, 1024, 256),
                                   (100352, 256, 64), (512, 128, 64)])
def test_gemm_nt_v6_matches_v1(shape, span):
    """v6 (256x128 tile, 8 waves; optional barrier-span glds) must be
    bitwise v1 on full-tile shapes."""
    M, N, K = shape
    torch.manual_seed(6)
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C1 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    C6 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    ext().gemm_nt_bf16(A, B, C1)
    ext().gemm_nt_bf16_v6(A, B, C6, span=span)
    torch.cuda.synchronize()
    assert torch.equal(C1, C6), (
        span, (C1.float() - C6.float()).abs().max().item()
    )


def test_gemm_v6_throughput_readout():
    M, N, K = 25088, 512, 512
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)

    def t(f, n=30):
        for _ in range(5):
            f()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            f()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    flops = 2.0 * M * N * K
    t5 = t(lambda: ext().gemm_nt_bf16_v5(A, B, C))
    t6 = t(lambda: ext().gemm_nt_bf16_v6(A, B, C, span=False))
    t6s = t(lambda: ext().gemm_nt_bf16_v6(A, B, C, span=True))
    print(f"\n[gemm v5 vs v6 vs v6-span] {flops / t5 / 1e12:.1f} vs "
          f"{flops / t6 / 1e12:.1f} vs {flops / t6s / 1e12:.1f} TF")


@pytest.mark.parametrize("shape,split", [
    ((1568, 512, 1024), 8), ((1568, 2048, 512), 4),
    ((640, 128, 512), 2), ((130, 72, 512), 3),
])
def test_gemm_nt_splitk_matches_matmul(shape, split):
    M, N, K = shape
    torch.manual_seed(7)
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    ext().gemm_nt_splitk_bf16(A, B, C, split)
    torch.cuda.synchronize()
    ref = A.float() @ B.float().t()
    err = (C.float() - ref).abs()
    scale = ref.abs().mean() + 1e-3
    assert (err.mean() / scale) < 5e-2, (err.mean() / scale).item()
    assert torch.allclose(C.float(), ref, atol=3.0, rtol=8e-2), (
        err.max().item()
    )


def test_gemm_splitk_throughput_readout():
    M, N, K = 1568, 512, 1024
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)

    def t(f, n=30):
        for _ in range(5):
            f()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            f()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    flops = 2.0 * M * N * K
    t2 = t(lambda: ext().gemm_nt_bf16_v2(A, B, C))
    tk = t(lambda: ext().gemm_nt_splitk_bf16(A, B, C, 8))
    print(f"\n[gemm 1568x512x1024 v2 vs splitk8] {flops / t2 / 1e12:.1f} "
          f"vs {flops / tk / 1e12:.1f} TF")


def test_gemm_nt_v4_matches_v1_and_throughput():
    M, N, K = 25088, 512, 512
    torch.manual_seed(4)
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C1 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    C4 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    ext().gemm_nt_bf16(A, B, C1)
    ext().gemm_nt_bf16_v4(A, B, C4)
    torch.cuda.synchronize()
    assert torch.equal(C1, C4), (
        (C1.float() - C4.float()).abs().max().item()
    )

    def t(f, n=30):
        for _ in range(5):
            f()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            f()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    flops = 2.0 * M * N * K
    t3 = t(lambda: ext().gemm_nt_bf16_v3(A, B, C4))
    t4 = t(lambda: ext().gemm_nt_bf16_v4(A, B, C4))
    print(f"\n[gemm v3 vs v4-span] {flops / t3 / 1e12:.1f} TF vs "
          f"{flops / t4 / 1e12:.1f} TF")


def test_mfma_conv1x1_matches_conv2d():
    """Full conv module on the hand-written MFMA GEMMs vs nn.Conv2d,
    fwd + dgrad + wgrad, stride 1 and 2."""
    import torch.nn.functional as F

    from stochastic_gradient_push_amd.models.layers import MfmaConv1x1

    CL = torch.channels_last
    for stride, cin, cout in ((1, 64, 128), (2, 256, 512), (1, 192, 96)):
        torch.manual_seed(0)
        ref = torch.nn.Conv2d(cin, cout, 1, stride=stride, bias=False).to(
            dev()
        )
        m = MfmaConv1x1(cin, cout, stride=stride).to(dev())
        m.weight.data.copy_(ref.weight.data)

        x = (
            torch.randn(4, cin, 14, 14, device=dev())
            .to(torch.bfloat16)
            .contiguous(memory_format=CL)
            .requires_grad_(True)
        )
        x2 = x.detach().clone().requires_grad_(True)
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
            y1 = m(x)
            y2 = ref(x2)
        assert torch.allclose(
            y1.float(), y2.float(), atol=5e-2, rtol=5e-2
        ), (stride, cin, cout, (y1.float() - y2.float()).abs().max())
        dy = torch.randn_like(y1)
        y1.backward(dy)
        y2.backward(dy)
        torch.cuda.synchronize()
        assert torch.allclose(
            x.grad.float(), x2.grad.float(), atol=8e-2, rtol=8e-2
        ), (stride, (x.grad.float() - x2.grad.float()).abs().max())
        cos = F.cosine_similarity(
            m.weight.grad.flatten(), ref.weight.grad.flatten(), dim=0
        )
        assert cos.item() > 0.999, cos.item()


def test_mfma_resnet_block_forward_backward():
    """Bottleneck block with conv_impl='mfma' trains (finite) on GPU."""
    from stochastic_gradient_push_amd.models import build_resnet

    m = build_resnet(
        "resnet50", num_classes=10, norm="fused", conv_impl="mfma"
    ).to(dev()).to(memory_format=torch.channels_last)
    x = torch.randn(2, 3, 64, 64, device=dev()).contiguous(
        memory_format=torch.channels_last
    )
    y = torch.randint(0, 10, (2,), device=dev())
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        loss = torch.nn.functional.cross_entropy(m(x), y)
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item()


@pytest.mark.parametrize("span", [False, True])
def test_gemm_nt_v7_matches_v1(span):
    """v7 (LDS-staged vectorized epilogue) must be bitwise v1."""
    M, N, K = 25088, 512, 512
    torch.manual_seed(8)
    A = torch.randn(M, K, device=dev()).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev()).to(torch.bfloat16)
    C1 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    C7 = torch.zeros(M, N, device=dev(), dtype=torch.bfloat16)
    ext().gemm_nt_bf16(A, B, C1)
    ext().gemm_nt_bf16_v7(A, B, C7, span=span)
    torch.cuda.synchronize()
    assert torch.equal(C1, C7), (
        span, (C1.float() - C7.float()).abs().max().item()
    )
