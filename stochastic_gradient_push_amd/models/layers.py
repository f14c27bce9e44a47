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

import os

import torch
import torch.nn as nn

# one-launch BN reduce+finalize (agent-scope sc1 partial stores +
# last-block counter).  v1 (device fences) measured 3x slower; v2 is
# gated here until measured.
_BN_FUSE = os.environ.get("SGP_BN_FUSE", "0") == "1"


class NativeBatchNorm2d(nn.BatchNorm2d):
    """BatchNorm2d that always uses torch's native kernels (not MIOpen)
    and no num_batches_tracked counter."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, affine=True):
        super().__init__(
            num_features, eps=eps, momentum=momentum, affine=affine,
            track_running_stats=True,
        )
        # the counter buffer stays registered (so checkpoints remain
        # interchangeable with stock nn.BatchNorm2d under strict
        # load_state_dict) but is never incremented: forward() calls
        # torch.batch_norm directly with a constant momentum, skipping
        # the per-BN long-add kernel the stock forward launches

    def forward(self, x):
        momentum = self.momentum
        if momentum is None:
            # cumulative moving average — the one case that needs the
            # counter (and it's updated host-side, no extra kernel)
            if self.training:
                self.num_batches_tracked += 1
            momentum = 1.0 / float(self.num_batches_tracked.clamp(min=1))
        return torch.batch_norm(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            self.training, momentum, self.eps,
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


class GemmConv1x1(nn.Conv2d):
    """1x1 convolution dispatched as a plain GEMM.

    In channels_last a 1x1 conv IS ``y[M, Co] = x[M, Ci] @ W[Co, Ci]^T``
    with M = N*H*W and zero-copy reshapes; ``torch.matmul`` routes it to
    hipBLASLt's tuned bf16 MFMA GEMMs instead of MIOpen's conv kernels
    (profiled far below the MFMA roofline at these shapes).  Stride-2
    variants subsample rows first.  Autograd covers dgrad/wgrad as the
    transposed GEMMs through the same library path.
    """

    def __init__(self, cin, cout, stride=1):
        super().__init__(cin, cout, 1, stride=stride, bias=False)

    def forward(self, x):
        if not (
            x.is_cuda
            and x.dtype in (torch.bfloat16, torch.float16)
            and x.is_contiguous(memory_format=torch.channels_last)
        ):
            return super().forward(x)
        if self.stride[0] != 1:
            x = x[:, :, :: self.stride[0], :: self.stride[1]]
            x = x.contiguous(memory_format=torch.channels_last)
        n, c, h, w = x.shape
        x2d = x.permute(0, 2, 3, 1).reshape(n * h * w, c)  # zero-copy
        weight = self.weight.view(self.out_channels, c).to(x.dtype)
        y2d = torch.matmul(x2d, weight.t())
        return (
            y2d.view(n, h, w, self.out_channels).permute(0, 3, 1, 2)
        )  # channels_last strides, zero-copy


class _MfmaConv1x1Fn(torch.autograd.Function):
    """1x1 conv entirely on the hand-written MFMA GEMMs
    (ops/csrc/gemm1x1_kernels.hip, hardware-probe-verified):
      fwd   y[M,Co] = x[M,Ci] @ W[Co,Ci]^T          (NT kernel)
      dgrad dx[M,Ci] = dy[M,Co] @ W = dy @ (W^T)^T  (NT kernel, W^T)
      wgrad dW[Co,Ci] = dy^T @ x                    (split-M TN kernel
            with deterministic partial reduce, hardware-validated)
    """

    @staticmethod
    def forward(ctx, x2d, weight):
        from .. import ops as _o

        w_bf16 = weight.detach().to(torch.bfloat16)
        y = _o.gemm_nt(x2d, w_bf16)
        ctx.save_for_backward(x2d, w_bf16)
        return y

    @staticmethod
    def backward(ctx, dy):
        from .. import ops as _o

        x2d, w_bf16 = ctx.saved_tensors
        dy = dy.contiguous()
        wt = w_bf16.t().contiguous()  # [Ci, Co]
        dx = _o.gemm_nt(dy, wt)
        dw = _o.gemm_tn_wgrad(dy, x2d)
        return dx, dw


class MfmaConv1x1(nn.Conv2d):
    """1x1 convolution on the hand-written MFMA GEMM kernels (north-star
    conv path; ``conv_impl='mfma'``/'auto').  The NT/TN ladder beats
    hipBLASLt on several ResNet shapes (profiles/r02_summary.md);
    MIOpen stays the measured end-to-end default because its in-graph
    wgrad/fwd mix is still faster in aggregate (r02 steady-state
    analysis)."""

    def __init__(self, cin, cout, stride=1):
        super().__init__(cin, cout, 1, stride=stride, bias=False)

    def forward(self, x):
        if not (
            x.is_cuda
            and x.dtype == torch.bfloat16
            and x.is_contiguous(memory_format=torch.channels_last)
            and self.in_channels % 32 == 0
        ):
            return super().forward(x)
        strided = self.stride[0] != 1
        x_in = x
        if strided:
            x_in = x[:, :, :: self.stride[0], :: self.stride[1]].contiguous(
                memory_format=torch.channels_last
            )
        n, c, h, w = x_in.shape
        x2d = x_in.permute(0, 2, 3, 1).reshape(n * h * w, c)
        y2d = _MfmaConv1x1Fn.apply(x2d.contiguous(), self.weight.view(
            self.out_channels, c
        ))
        return y2d.view(n, h, w, self.out_channels).permute(0, 3, 1, 2)


class _MfmaConv3x3Fn(torch.autograd.Function):
    """3x3 pad-1 conv on the hand-written MFMA implicit-GEMM kernel
    (ops/csrc/conv3x3_kernels.hip).

    fwd:   implicit-GEMM kernel (stride 1 or 2)
    dgrad: stride 1 -> the SAME kernel on dy with the rotated/transposed
           weight w_rot[ci,r',s',co] = w[co,2-r',2-s',ci];
           stride 2 -> torch.nn.grad.conv2d_input (MIOpen fallback)
    wgrad: hand-written implicit-TN kernel (split-M partials +
           deterministic reduce), both strides
    """

    @staticmethod
    def forward(ctx, x, weight, stride):
        from .. import ops as _o

        k = _o._ext_for(x)
        n, ci, h, w = x.shape
        co = weight.shape[0]
        ho = (h - 1) // stride + 1
        wo = (w - 1) // stride + 1
        # [Co,3,3,Ci] contiguous = channels_last view of the weight
        w_bf = (
            weight.detach().permute(0, 2, 3, 1).contiguous()
            .to(torch.bfloat16)
        )
        y = torch.empty(
            n, co, ho, wo, device=x.device, dtype=torch.bfloat16,
            memory_format=torch.channels_last,
        )
        k.conv3x3_nhwc_bf16(x, w_bf, y, stride)
        ctx.save_for_backward(x, w_bf)
        ctx.c3_stride = stride
        ctx.c3_wdtype = weight.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        from .. import ops as _o

        x, w_bf = ctx.saved_tensors
        stride = ctx.c3_stride
        dy = dy.contiguous(memory_format=torch.channels_last)
        k = _o._ext_for(x)
        # MIOpen fallbacks need a CONTIGUOUS NCHW weight: a permuted
        # strided view makes conv2d_input/conv2d_weight fall back to the
        # naive double-precision kernels (measured 30 ms per call)
        w_nchw = w_bf.permute(0, 3, 1, 2).contiguous()
        if stride == 1:
            w_rot = w_bf.flip(1, 2).permute(3, 1, 2, 0).contiguous()
            dx = torch.empty_like(x)
            k.conv3x3_nhwc_bf16(dy, w_rot, dx, 1)
        else:
            dx = torch.nn.grad.conv2d_input(
                list(x.shape), w_nchw, dy, stride=stride, padding=1,
            )
        co, ci = w_bf.shape[0], w_bf.shape[3]
        dw_flat = torch.empty(
            co * 9 * ci, device=x.device, dtype=torch.float32
        )
        k.conv3x3_wgrad_bf16(x, dy, dw_flat, stride)
        # [Co*3*3*Ci] fp32 -> channels_last-strided [Co,Ci,3,3] grad
        dw = dw_flat.view(co, 3, 3, ci).permute(0, 3, 1, 2).to(
            ctx.c3_wdtype
        )
        return dx, dw, None


class MfmaConv3x3(nn.Conv2d):
    """3x3 convolution on the hand-written MFMA implicit-GEMM kernel
    (``conv_impl='mfma'``); falls back to the stock conv for inputs the
    kernel doesn't cover (CPU, fp32, non-channels-last, Ci % 64 != 0)."""

    def __init__(self, cin, cout, stride=1):
        super().__init__(cin, cout, 3, stride=stride, padding=1, bias=False)

    def forward(self, x):
        if not (
            x.is_cuda
            and x.dtype == torch.bfloat16
            and x.is_contiguous(memory_format=torch.channels_last)
            and self.in_channels % 64 == 0
        ):
            return super().forward(x)
        return _MfmaConv3x3Fn.apply(x, self.weight, self.stride[0])


def _fused_supported(x: torch.Tensor, C: int) -> bool:
    """The hand-written kernels need bf16 NHWC with C = 8 * 2^k <= 2048
    (the reduce kernel's thread geometry; every ResNet width qualifies)."""
    cdiv8 = C // 8
    return (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and C % 8 == 0
        and cdiv8 <= 256
        and (cdiv8 & (cdiv8 - 1)) == 0
        and x.is_contiguous(memory_format=torch.channels_last)
    )


class _FusedBNFunction(torch.autograd.Function):
    """Fused NHWC BN(+residual)(+ReLU), bf16 activations / fp32 stats.

    Forward: 1 reduce + 1 finalize + 1 apply kernel (training).
    Backward: 1 reduce (dgamma/dbeta) + 1 apply (dx [, dresidual]).
    """

    @staticmethod
    def forward(ctx, x, residual, gamma, beta, running_mean, running_var,
                momentum, eps, training, relu, counter=None):
        from .. import ops as _o

        k = _o._ext_for(x)
        N, C, H, W = x.shape
        M = N * H * W
        y = torch.empty_like(x)
        scale_shift = torch.empty(2 * C, device=x.device, dtype=torch.float32)
        if training:
            scratch = torch.empty(
                k.bn_partials_numel(M, C), device=x.device,
                dtype=torch.float32,
            )
            smean = torch.empty(C, device=x.device, dtype=torch.float32)
            sinvstd = torch.empty(C, device=x.device, dtype=torch.float32)
            if counter is not None:
                # reduce + last-block finalize in ONE launch
                k.bn_fwd_reduce_finalize(
                    x, scratch, counter, gamma, beta, running_mean,
                    running_var, smean, sinvstd, scale_shift, momentum,
                    eps, M, C, True,
                )
            else:
                k.bn_fwd_reduce(x, scratch, M, C)
                k.bn_fwd_finalize(
                    scratch, gamma, beta, running_mean, running_var, smean,
                    sinvstd, scale_shift, momentum, eps, M, C, True,
                )
        else:
            k.bn_eval_prep(running_mean, running_var, gamma, beta,
                           scale_shift, eps, C)
            smean = running_mean
            sinvstd = torch.rsqrt(running_var + eps)
        k.bn_fwd_apply(x, residual, y, scale_shift, M, C, relu)
        ctx.save_for_backward(x, y, gamma, smean, sinvstd)
        ctx.bn_shape = (M, C)
        ctx.bn_relu = relu
        ctx.bn_training = training
        ctx.bn_has_res = residual is not None
        ctx.bn_counter = counter
        return y

    @staticmethod
    def backward(ctx, dy):
        from .. import ops as _o

        x, y, gamma, smean, sinvstd = ctx.saved_tensors
        k = _o._ext_for(x)
        M, C = ctx.bn_shape
        relu = ctx.bn_relu
        dy = dy.contiguous(memory_format=torch.channels_last)
        scratch = torch.empty(
            k.bn_partials_numel(M, C), device=x.device, dtype=torch.float32
        )
        dgamma = torch.empty(C, device=x.device, dtype=torch.float32)
        dbeta = torch.empty(C, device=x.device, dtype=torch.float32)
        coef = torch.empty(3 * C, device=x.device, dtype=torch.float32)
        counter = ctx.bn_counter
        if counter is not None and ctx.bn_training:
            k.bn_bwd_reduce_finalize(
                x, dy, y if relu else None, smean, sinvstd, scratch,
                counter, gamma, dgamma, dbeta, coef, M, C, relu,
                ctx.bn_training,
            )
        else:
            k.bn_bwd_reduce(x, dy, y if relu else None, smean, sinvstd,
                            scratch, M, C, relu)
            k.bn_bwd_finalize(scratch, gamma, smean, sinvstd, dgamma,
                              dbeta, coef, M, C, ctx.bn_training)
        dx = torch.empty_like(x)
        dres = torch.empty_like(x) if ctx.bn_has_res else None
        k.bn_bwd_apply(x, dy, y if relu else None, dx, dres, coef, M, C, relu)
        return (dx, dres, dgamma, dbeta, None, None, None, None, None, None,
                None)


class FusedBatchNorm2d(nn.BatchNorm2d):
    """BatchNorm2d[+residual add][+ReLU] in hand-written CDNA4 kernels.

    ``forward(x, residual=None)``: when ``residual`` is given it is added
    after normalization, before the (optional) ReLU — the ResNet residual
    join.  Falls back to the equivalent PyTorch ops for inputs the
    kernels don't cover (CPU, fp32, non-channels-last).
    """

    def __init__(self, num_features, eps=1e-5, momentum=0.1, relu=False):
        super().__init__(
            num_features, eps=eps, momentum=momentum, affine=True,
            track_running_stats=True,
        )
        self.relu = relu
        # last-block-finalize counter (device int32, lazily created; the
        # fused reduce+finalize kernel resets it after each use)
        self._bn_ctr = None
        # counter kept (checkpoint interchange with nn.BatchNorm2d) but
        # never incremented — see NativeBatchNorm2d

    def forward(self, x, residual=None):
        if _fused_supported(x, self.num_features) and (
            residual is None
            or residual.is_contiguous(memory_format=torch.channels_last)
        ):
            if _BN_FUSE and (
                self._bn_ctr is None or self._bn_ctr.device != x.device
            ):
                self._bn_ctr = torch.zeros(
                    1, dtype=torch.int32, device=x.device
                )
            with torch.amp.autocast(device_type="cuda", enabled=False):
                return _FusedBNFunction.apply(
                    x, residual, self.weight, self.bias, self.running_mean,
                    self.running_var, self.momentum, self.eps,
                    self.training, self.relu, self._bn_ctr,
                )
        # reference fallback path (also the CPU numerics oracle)
        y = torch.batch_norm(
            x, self.weight, self.bias, self.running_mean, self.running_var,
            self.training, self.momentum, self.eps, False,
        )
        if residual is not None:
            y = y + residual
        if self.relu:
            y = torch.relu(y)
        return y
