#!/usr/bin/env python3
"""Tight loops over the round-2 MFMA kernels for rocprofv3 --pmc runs.

    rocprofv3 --kernel-trace --stats \
        --pmc SQ_VALU_MFMA_BUSY_CYCLES SQ_BUSY_CYCLES \
              SQ_LDS_BANK_CONFLICT SQ_WAIT_ANY \
        -d OUT -- python tools/kernels_pmc.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from stochastic_gradient_push_amd import ops  # noqa: E402


def main():
    dev = torch.device("cuda", 0)
    ext = ops._ext_for(torch.empty(1, device=dev))
    torch.manual_seed(0)

    # v6 NT GEMM (the headline 1x1 shape)
    M, N, K = 25088, 512, 512
    A = torch.randn(M, K, device=dev).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev).to(torch.bfloat16)
    C = torch.zeros(M, N, device=dev, dtype=torch.bfloat16)
    for _ in range(20):
        ext.gemm_nt_bf16_v6(A, B, C, span=True)

    # wgrad TN (double-buffered)
    dw = torch.zeros(N * K, device=dev)
    for _ in range(20):
        ops.gemm_tn_wgrad(C, A)

    # conv3x3 fwd (28x28 stage)
    x = torch.randn(32, 128, 28, 28, device=dev).to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    w = torch.randn(128, 3, 3, 128, device=dev).to(torch.bfloat16)
    y = torch.empty(32, 128, 28, 28, device=dev, dtype=torch.bfloat16,
                    memory_format=torch.channels_last)
    for _ in range(20):
        ext.conv3x3_nhwc_bf16(x, w.contiguous(), y, 1)
    torch.cuda.synchronize()
    print("pmc loops done", dw.shape)


if __name__ == "__main__":
    main()
