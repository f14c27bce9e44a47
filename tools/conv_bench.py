#!/usr/bin/env python3
"""Per-shape fwd+bwd timing: MfmaConv3x3/MfmaConv1x1 vs stock (MIOpen)
conv on every distinct ResNet-50 conv shape at bs=32.  Drives the
conv_impl='auto' per-shape selection."""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from stochastic_gradient_push_amd.models.layers import (  # noqa: E402
    MfmaConv1x1,
    MfmaConv3x3,
)

CL = torch.channels_last


def timeit(f, n=20, warm=5):
    for _ in range(warm):
        f()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        f()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def bench_pair(make_ours, make_ref, x_shape):
    dev = torch.device("cuda", 0)
    ours = make_ours().to(dev)
    ref = make_ref().to(dev).to(memory_format=CL)
    ref.weight.data.copy_(ours.weight.data)

    def run(m, autocast):
        x = torch.randn(*x_shape, device=dev).to(torch.bfloat16)
        x = x.contiguous(memory_format=CL).requires_grad_(True)

        def step():
            if autocast:
                with torch.autocast(device_type="cuda",
                                    dtype=torch.bfloat16):
                    y = m(x)
            else:
                y = m(x)
            y.backward(torch.ones_like(y))
            x.grad = None
            m.weight.grad = None

        return timeit(step)

    return run(ours, False), run(ref, True)


def main():
    torch.backends.cudnn.benchmark = True
    bs = 32
    print("3x3 convs (fwd+bwd ms): ours vs miopen")
    for ci, h, co, stride in [
        (64, 56, 64, 1), (128, 56, 128, 2), (128, 28, 128, 1),
        (256, 28, 256, 2), (256, 14, 256, 1), (512, 14, 512, 2),
        (512, 7, 512, 1),
    ]:
        t_o, t_r = bench_pair(
            lambda ci=ci, co=co, s=stride: MfmaConv3x3(ci, co, stride=s),
            lambda ci=ci, co=co, s=stride: torch.nn.Conv2d(
                ci, co, 3, stride=s, padding=1, bias=False
            ),
            (bs, ci, h, h),
        )
        win = "OURS" if t_o < t_r else "miopen"
        print(f"  3x3 {ci:>4}x{h}x{h} s{stride} -> {co:>4}: "
              f"{t_o * 1e3:7.3f} vs {t_r * 1e3:7.3f} ms  [{win}]")

    print("1x1 convs (fwd+bwd ms): ours vs miopen")
    for ci, h, co, stride in [
        (64, 56, 64, 1), (64, 56, 256, 1), (256, 56, 64, 1),
        (256, 56, 128, 1), (256, 56, 512, 2), (128, 28, 512, 1),
        (512, 28, 128, 1), (512, 28, 256, 1), (512, 28, 1024, 2),
        (256, 14, 1024, 1), (1024, 14, 256, 1), (1024, 14, 512, 1),
        (1024, 14, 2048, 2), (512, 7, 2048, 1), (2048, 7, 512, 1),
    ]:
        t_o, t_r = bench_pair(
            lambda ci=ci, co=co, s=stride: MfmaConv1x1(ci, co, stride=s),
            lambda ci=ci, co=co, s=stride: torch.nn.Conv2d(
                ci, co, 1, stride=s, bias=False
            ),
            (bs, ci, h, h),
        )
        win = "OURS" if t_o < t_r else "miopen"
        print(f"  1x1 {ci:>4}x{h}x{h} s{stride} -> {co:>4}: "
              f"{t_o * 1e3:7.3f} vs {t_r * 1e3:7.3f} ms  [{win}]")


if __name__ == "__main__":
    main()
