#!/usr/bin/env python3
"""MFMA GEMM ladder benchmark on every ResNet-50 1x1 shape (bs=32).

Times v1/v2/v3/v5 NT kernels and the wgrad TN kernel against
torch.matmul (hipBLASLt), prints a TF table.  Run on the GPU box:
    python tools/gemm_bench.py [--bs 32]
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from stochastic_gradient_push_amd import ops  # noqa: E402


def timeit(f, n=30, warm=5):
    for _ in range(warm):
        f()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        f()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--bs", type=int, default=32)
    p.add_argument("--wgrad-split", type=int, default=8)
    args = p.parse_args()
    dev = torch.device("cuda", 0)
    ext = ops._ext_for(torch.empty(1, device=dev))

    # (M, N, K) of every distinct 1x1-conv GEMM in ResNet-50 fwd at bs
    b = args.bs
    shapes = [
        (b * 56 * 56, 64, 64),
        (b * 56 * 56, 64, 256),
        (b * 56 * 56, 256, 64),
        (b * 28 * 28, 128, 256),   # stride-2 projection input subsampled
        (b * 28 * 28, 128, 512),
        (b * 28 * 28, 512, 128),
        (b * 28 * 28, 512, 256),
        (b * 14 * 14, 256, 512),
        (b * 14 * 14, 256, 1024),
        (b * 14 * 14, 1024, 256),
        (b * 14 * 14, 1024, 512),
        (b * 7 * 7, 512, 1024),
        (b * 7 * 7, 512, 2048),
        (b * 7 * 7, 2048, 512),
        (b * 7 * 7, 2048, 1024),
        (25088, 512, 512),         # the ladder's reference shape
    ]

    cols = ("v2", "v3", "v5", "v6", "v6s", "v7", "auto", "lib")
    print(f"{'M':>8} {'N':>5} {'K':>5} | " + " ".join(
        f"{c:>6}" for c in cols) + "  (TFLOP/s)")
    for M, N, K in shapes:
        A = torch.randn(M, K, device=dev).to(torch.bfloat16)
        B = torch.randn(N, K, device=dev).to(torch.bfloat16)
        C = torch.zeros(M, N, device=dev, dtype=torch.bfloat16)
        fl = 2.0 * M * N * K
        r = dict.fromkeys(cols, 0.0)
        r["v2"] = fl / timeit(lambda: ext.gemm_nt_bf16_v2(A, B, C)) / 1e12
        if K % 64 == 0:
            r["v3"] = fl / timeit(
                lambda: ext.gemm_nt_bf16_v3(A, B, C)) / 1e12
            r["v5"] = fl / timeit(
                lambda: ext.gemm_nt_bf16_v5(A, B, C)) / 1e12
            if M % 256 == 0 and N % 128 == 0:
                r["v6"] = fl / timeit(lambda: ext.gemm_nt_bf16_v6(
                    A, B, C, span=False)) / 1e12
                r["v6s"] = fl / timeit(lambda: ext.gemm_nt_bf16_v6(
                    A, B, C, span=True)) / 1e12
                r["v7"] = fl / timeit(lambda: ext.gemm_nt_bf16_v7(
                    A, B, C, span=True)) / 1e12
        r["auto"] = fl / timeit(lambda: ops.gemm_nt(A, B, out=C)) / 1e12
        r["lib"] = fl / timeit(lambda: torch.matmul(A, B.t())) / 1e12
        print(f"{M:>8} {N:>5} {K:>5} | " + " ".join(
            f"{r[k]:>6.0f}" for k in cols))

    # wgrad TN: dW[Co,Ci] = dy^T @ x on the same shapes (Co=N, Ci=K);
    # "tn" uses the dispatcher's split heuristic
    print("\nwgrad TN (dispatcher split):")
    print(f"{'M':>8} {'Co':>5} {'Ci':>5} | {'tn':>6} {'lib':>6}  (TFLOP/s)")
    for M, Co, Ci in shapes:
        dy = torch.randn(M, Co, device=dev).to(torch.bfloat16)
        x = torch.randn(M, Ci, device=dev).to(torch.bfloat16)
        fl = 2.0 * M * Co * Ci
        t_tn = fl / timeit(lambda: ops.gemm_tn_wgrad(dy, x)) / 1e12
        t_lib = fl / timeit(lambda: torch.matmul(dy.t(), x)) / 1e12
        print(f"{M:>8} {Co:>5} {Ci:>5} | {t_tn:>6.0f} {t_lib:>6.0f}")


if __name__ == "__main__":
    main()
