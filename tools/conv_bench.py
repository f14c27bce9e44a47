#!/usr/bin/env python3
"""Per-shape per-primitive timing: our MFMA conv kernels vs MIOpen on
every distinct ResNet-50 conv shape at bs=32 (fwd / dgrad / wgrad
separately, tight launch loops so host overhead is comparable on both
sides).  Drives the conv_impl='auto' per-shape selection."""

import os
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from stochastic_gradient_push_amd import ops  # noqa: E402

CL = torch.channels_last


def timeit(f, n=30, warm=5):
    for _ in range(warm):
        f()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        f()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def bench_3x3(bs=32):
    dev = torch.device("cuda", 0)
    ext = ops._ext_for(torch.empty(1, device=dev))
    print("3x3 (us): fwd ours/mi | dgrad ours/mi | wgrad mi-only | "
          "total ours/mi")
    for ci, h, co, stride in [
        (64, 56, 64, 1), (128, 56, 128, 2), (128, 28, 128, 1),
        (256, 28, 256, 2), (256, 14, 256, 1), (512, 14, 512, 2),
        (512, 7, 512, 1),
    ]:
        x = torch.randn(bs, ci, h, h, device=dev).to(torch.bfloat16)
        x = x.contiguous(memory_format=CL)
        w = torch.randn(co, ci, 3, 3, device=dev).to(torch.bfloat16)
        ho = (h - 1) // stride + 1
        y = torch.empty(bs, co, ho, ho, device=dev, dtype=torch.bfloat16,
                        memory_format=CL)
        wp = w.permute(0, 2, 3, 1).contiguous()
        dy = torch.randn_like(y).contiguous(memory_format=CL)
        dx = torch.empty_like(x)
        w_rot = wp.flip(1, 2).permute(3, 1, 2, 0).contiguous()

        tf_o = timeit(lambda: ext.conv3x3_nhwc_bf16(x, wp, y, stride))
        tf_m = timeit(lambda: F.conv2d(x, w, stride=stride, padding=1))
        if stride == 1:
            td_o = timeit(lambda: ext.conv3x3_nhwc_bf16(dy, w_rot, dx, 1))
        else:
            td_o = None
        td_m = timeit(lambda: torch.nn.grad.conv2d_input(
            list(x.shape), w, dy, stride=stride, padding=1))
        tw_m = timeit(lambda: torch.nn.grad.conv2d_weight(
            x, list(w.shape), dy, stride=stride, padding=1))
        tot_o = tf_o + (td_o if td_o is not None else td_m) + tw_m
        tot_m = tf_m + td_m + tw_m
        win = "OURS" if tot_o < tot_m else "mi"
        d_o = f"{td_o * 1e6:5.0f}" if td_o is not None else " mi  "
        print(f"  {ci:>4}x{h:>2} s{stride} -> {co:>4}: "
              f"{tf_o * 1e6:5.0f}/{tf_m * 1e6:5.0f} | "
              f"{d_o}/{td_m * 1e6:5.0f} | {tw_m * 1e6:5.0f} | "
              f"{tot_o * 1e6:5.0f}/{tot_m * 1e6:5.0f} [{win}]")


def bench_1x1(bs=32):
    dev = torch.device("cuda", 0)
    print("1x1 (us): fwd ours/mi | dgrad ours/mi | wgrad ours/mi | "
          "total ours/mi")
    for ci, h, co, stride in [
        (64, 56, 64, 1), (64, 56, 256, 1), (256, 56, 64, 1),
        (256, 56, 128, 1), (256, 56, 512, 2), (128, 28, 512, 1),
        (512, 28, 128, 1), (512, 28, 256, 1), (512, 28, 1024, 2),
        (256, 14, 1024, 1), (1024, 14, 256, 1), (1024, 14, 512, 1),
        (1024, 14, 2048, 2), (512, 7, 2048, 1), (2048, 7, 512, 1),
    ]:
        x = torch.randn(bs, ci, h, h, device=dev).to(torch.bfloat16)
        x = x.contiguous(memory_format=CL)
        w = torch.randn(co, ci, 1, 1, device=dev).to(torch.bfloat16) * 0.05
        ho = (h - 1) // stride + 1
        # GEMM views (what MfmaConv1x1 does)
        if stride == 1:
            x2d = x.permute(0, 2, 3, 1).reshape(-1, ci).contiguous()
        else:
            xs = x[:, :, ::stride, ::stride].contiguous(memory_format=CL)
            x2d = xs.permute(0, 2, 3, 1).reshape(-1, ci).contiguous()
        w2d = w.view(co, ci)
        wt = w2d.t().contiguous()
        M = bs * ho * ho
        dy2d = torch.randn(M, co, device=dev).to(torch.bfloat16)
        dy4d = (
            dy2d.view(bs, ho, ho, co).permute(0, 3, 1, 2)
            .contiguous(memory_format=CL)
        )

        tf_o = timeit(lambda: ops.gemm_nt(x2d, w2d))
        tf_m = timeit(lambda: F.conv2d(x, w, stride=stride))
        td_o = timeit(lambda: ops.gemm_nt(dy2d, wt))
        td_m = timeit(lambda: torch.nn.grad.conv2d_input(
            list(x.shape), w, dy4d, stride=stride))
        tw_o = timeit(lambda: ops.gemm_tn_wgrad(dy2d, x2d))
        tw_m = timeit(lambda: torch.nn.grad.conv2d_weight(
            x, list(w.shape), dy4d, stride=stride))
        tot_o, tot_m = tf_o + td_o + tw_o, tf_m + td_m + tw_m
        win = "OURS" if tot_o < tot_m else "mi"
        print(f"  {ci:>4}x{h:>2} s{stride} -> {co:>4}: "
              f"{tf_o * 1e6:5.0f}/{tf_m * 1e6:5.0f} | "
              f"{td_o * 1e6:5.0f}/{td_m * 1e6:5.0f} | "
              f"{tw_o * 1e6:5.0f}/{tw_m * 1e6:5.0f} | "
              f"{tot_o * 1e6:5.0f}/{tot_m * 1e6:5.0f} [{win}]")


if __name__ == "__main__":
    torch.backends.cudnn.benchmark = True
    bench_3x3()
    bench_1x1()
