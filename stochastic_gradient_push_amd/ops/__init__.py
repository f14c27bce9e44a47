"""Fused flat-buffer device ops.

The gossip hot path of the reference is ~161 tiny per-tensor elementwise
CUDA ops per step (reference gossip/distributed.py:298-314, 372-379,
402-425; gossip/ad_psgd.py:357-361).  Here every hot op is a single fused
HIP/CDNA4 kernel launch over one contiguous flat buffer
(`csrc/gossip_kernels.hip`, built for gfx950):

==================  =====================================================
op                  semantics (all in-place over flat 1-D buffers)
==================  =====================================================
scale_              x *= a                  (bias / de-bias,
                                             reference distributed.py:302-313)
add_scale_          x = (x + r) * a         (residual accumulate + lazy
                                             mix, reference distributed.py:372-379)
pack_mix_           x *= a; out = x         (transfer: scale params and
                                             pack comm buffer in one pass,
                                             reference distributed.py:409-418)
average_            x = (x + y) * 0.5       (bilateral avg,
                                             reference ad_psgd.py:357-361)
sgd_step_           fused momentum-SGD update (reference relies on
                                             torch.optim.SGD, ad_psgd.py:261-266)
==================  =====================================================

Scalars ``a`` may be Python floats or 1-element device tensors (no host
sync on the training path: push-sum weights stay device-resident).

Dispatch: CUDA(HIP) tensors require the compiled ``_gossip_kernels``
extension — a missing extension on a GPU box raises immediately rather
than silently falling back to eager.  CPU tensors use the PyTorch
reference implementations (these are also the numerics oracle in tests).
"""

from typing import Optional, Union

import torch

Scalar = Union[float, torch.Tensor]

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from . import _gossip_kernels  # in-tree built .so

        _EXT = _gossip_kernels
    except ImportError as e:  # pragma: no cover - exercised on GPU boxes
        _EXT_ERR = str(e)
    return _EXT


def extension_available() -> bool:
    return _load_extension() is not None


def _ext_for(t: torch.Tensor):
    """Return the extension for a CUDA tensor, or raise loudly."""
    ext = _load_extension()
    if ext is None:
        raise RuntimeError(
            "stochastic_gradient_push_amd HIP extension (_gossip_kernels) is "
            "not built but a CUDA tensor reached the fused-op path. Build it "
            f"with `python setup.py build_ext --inplace`. Import error: {_EXT_ERR}"
        )
    return ext


def _as_scalar_tensor(a: Scalar, like: torch.Tensor) -> torch.Tensor:
    if isinstance(a, torch.Tensor):
        # cpu-comm mode keeps push-sum scalars on the comm device (CPU)
        # while the flat params are CUDA — move on mismatch (cold path;
        # device-comm mode passes device-resident scalars straight through)
        if a.device != like.device:
            a = a.to(like.device)
        if a.dtype != torch.float32:
            a = a.float()
        return a
    return torch.tensor([float(a)], device=like.device, dtype=torch.float32)


def scale_(x: torch.Tensor, a: Scalar) -> torch.Tensor:
    """x *= a (fused bias/de-bias over the whole flat parameter buffer)."""
    if x.is_cuda:
        _ext_for(x).scale_(x, _as_scalar_tensor(a, x))
    else:
        x.mul_(a if isinstance(a, float) else a.to(x.dtype))
    return x


def add_scale_(x: torch.Tensor, r: torch.Tensor, a: Scalar = 1.0) -> torch.Tensor:
    """x = (x + r) * a in a single pass (residual accumulate + lazy mix)."""
    if x.is_cuda:
        _ext_for(x).add_scale_(x, r, _as_scalar_tensor(a, x))
    else:
        x.add_(r)
        if not (isinstance(a, float) and a == 1.0):
            x.mul_(a if isinstance(a, float) else a.to(x.dtype))
    return x


def pack_mix_(x: torch.Tensor, out: torch.Tensor, a: Scalar = 1.0) -> torch.Tensor:
    """x *= a; out = x — one read of x, two writes (transfer_params path)."""
    if x.is_cuda:
        _ext_for(x).pack_mix_(x, out, _as_scalar_tensor(a, x))
    else:
        if not (isinstance(a, float) and a == 1.0):
            x.mul_(a if isinstance(a, float) else a.to(x.dtype))
        out.copy_(x)
    return out


def pack_mix_cast_(
    x: torch.Tensor, out: torch.Tensor, a: Scalar = 1.0
) -> torch.Tensor:
    """x *= a; out = x.to(out.dtype) — wire-format pack.  When ``out`` is
    bf16 the cast fuses into the pack pass (half the xGMI bytes per
    gossip message)."""
    if out.dtype == x.dtype:
        return pack_mix_(x, out, a)
    assert out.dtype == torch.bfloat16 and x.dtype == torch.float32
    if x.is_cuda:
        _ext_for(x).pack_mix_bf16_(x, out, _as_scalar_tensor(a, x))
    else:
        if not (isinstance(a, float) and a == 1.0):
            x.mul_(a if isinstance(a, float) else a.to(x.dtype))
        out.copy_(x)
    return out


def add_scale_cast_(
    x: torch.Tensor, r: torch.Tensor, a: Scalar = 1.0
) -> torch.Tensor:
    """x = (x + r.to(x.dtype)) * a — wire-format accumulate."""
    if r.dtype == x.dtype:
        return add_scale_(x, r, a)
    assert r.dtype == torch.bfloat16 and x.dtype == torch.float32
    if x.is_cuda:
        _ext_for(x).add_scale_bf16_(x, r, _as_scalar_tensor(a, x))
    else:
        x.add_(r.to(x.dtype))
        if not (isinstance(a, float) and a == 1.0):
            x.mul_(a if isinstance(a, float) else a.to(x.dtype))
    return x


def average_(x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """x = (x + y) * 0.5 (bilateral gossip merge)."""
    if x.is_cuda:
        _ext_for(x).average_(x, y)
    else:
        x.add_(y).mul_(0.5)
    return x


def sgd_step_(
    params: torch.Tensor,
    grads: torch.Tensor,
    momentum_buf: torch.Tensor,
    lr: Scalar,
    momentum: float = 0.0,
    weight_decay: float = 0.0,
    dampening: float = 0.0,
    nesterov: bool = False,
    first_step: bool = False,
) -> None:
    """Fused SGD-with-momentum update over flat param/grad buffers.

    Semantics match torch.optim.SGD:
      d = g + wd * p
      buf = momentum * buf + (1 - dampening) * d   (buf = d on first step)
      d = d + momentum * buf  (nesterov)  |  d = buf  (plain momentum)
      p -= lr * d
    """
    if params.is_cuda:
        _ext_for(params).sgd_step_(
            params, grads, momentum_buf, _as_scalar_tensor(lr, params),
            float(momentum), float(weight_decay), float(dampening),
            bool(nesterov), bool(first_step),
        )
        return
    lr = float(lr) if not isinstance(lr, torch.Tensor) else float(lr.item())
    d = grads
    if weight_decay != 0.0:
        d = d.add(params, alpha=weight_decay)
    if momentum != 0.0:
        if first_step:
            momentum_buf.copy_(d)
        else:
            momentum_buf.mul_(momentum).add_(d, alpha=1.0 - dampening)
        d = d.add(momentum_buf, alpha=momentum) if nesterov else momentum_buf
    params.add_(d, alpha=-lr)


# --------------------------------------------------------------- MFMA GEMM
# Hand-written NT/TN bf16 GEMM dispatch (ops/csrc/gemm1x1_kernels.hip).
# Variant choice per shape; see profiles/ for the measured ladder.

#: use the 3-buffer barrier-crossing glds variant of the 256x128 kernel
#: (measured across boxes: 2-buffer syncthreads wins or ties the span
#: variant at this occupancy — 526/523/527 vs 501/521/531 TF)
V6_SPAN = False


def gemm_nt(A: torch.Tensor, B: torch.Tensor,
            out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """C[M,N] = A[M,K] @ B[N,K]^T on the best hand-written MFMA kernel
    for the shape (bf16 in/out, fp32 accumulate)."""
    ext = _ext_for(A)
    M, K = A.shape
    N = B.shape[0]
    C = out if out is not None else torch.empty(
        M, N, device=A.device, dtype=torch.bfloat16
    )
    # Variant choice fitted to the measured ladder over every ResNet-50
    # 1x1 shape (profiles/r02_gemm_bench3.txt):
    #   * K not 64-aligned               -> v2 (only kernel without the
    #     K%64 constraint)
    #   * small M (layer4, M=1568): chip under-fill dominates -> split-K
    #     when K is deep enough to amortize the fp32 partial traffic,
    #     else register-staged v2 (glds pipelines lose at 13 m-tiles)
    #   * N=64 (layer1): half the 128-wide tile is waste; v2 degrades
    #     least
    #   * N>=256 with 256|M: the 256x128 v6 tile (more flops per staged
    #     byte) wins; otherwise the 128x128 glds v5
    if K % 64 != 0:
        ext.gemm_nt_bf16_v2(A, B, C)
        return C
    tiles = ((M + 127) // 128) * ((N + 127) // 128)
    if M <= 2048:
        split = 0
        if K >= 1024 and tiles <= 256:
            split = min(64, max(2, 512 // tiles), K // 128,
                        (64 << 20) // (M * N * 4))
        if split >= 2:
            ext.gemm_nt_splitk_bf16(A, B, C, split)
        else:
            ext.gemm_nt_bf16_v2(A, B, C)
    elif N < 128:
        ext.gemm_nt_bf16_v2(A, B, C)
    elif N >= 256 and M % 256 == 0 and N % 128 == 0:
        ext.gemm_nt_bf16_v6(A, B, C, span=V6_SPAN)
    else:
        ext.gemm_nt_bf16_v5(A, B, C)
    return C


def gemm_tn_wgrad(dy: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """dW[Co,Ci] = dy[M,Co]^T @ x[M,Ci] (fp32 out) via the split-M TN
    kernel with deterministic partial reduction."""
    ext = _ext_for(dy)
    M, Co = dy.shape
    Ci = x.shape[1]
    co_tiles = (Co + 127) // 128
    ci_tiles = (Ci + 127) // 128
    # split sweep (profiles: r02_wgrad_split_sweep) peaks at
    # tiles*split ~ 512 blocks (2/CU) and falls beyond (partial
    # traffic + barrier amortization); deep strided reduces were the
    # bottleneck at split 2048
    split = max(1, 512 // (co_tiles * ci_tiles))
    split = min(split, 64, max(1, M // 128))
    max_split_mem = (128 << 20) // (Co * Ci * 4)
    split = max(1, min(split, max_split_mem))
    partials = torch.empty(split * Co * Ci, device=dy.device,
                           dtype=torch.float32)
    dw = torch.empty(Co * Ci, device=dy.device, dtype=torch.float32)
    ext.gemm_tn_wgrad_bf16(dy, x, partials, dw, split)
    return dw.view(Co, Ci)


def cast_shadow_(master: torch.Tensor, shadow: torch.Tensor) -> None:
    """shadow = bf16(master) in one pass (working-weight refresh)."""
    if master.is_cuda:
        _ext_for(master).cast_shadow_(master, shadow)
    else:
        shadow.copy_(master)


def sgd_step_bf16gs_(
    params: torch.Tensor,
    grads_bf16: torch.Tensor,
    momentum_buf: torch.Tensor,
    shadow: torch.Tensor,
    lr,
    momentum: float = 0.0,
    weight_decay: float = 0.0,
    dampening: float = 0.0,
    nesterov: bool = False,
    first_step: bool = False,
) -> None:
    """Fused SGD over the fp32 master with bf16 gradients; refreshes the
    bf16 working-weight shadow in the same pass."""
    if params.is_cuda:
        _ext_for(params).sgd_step_bf16gs_(
            params, grads_bf16, momentum_buf, shadow,
            _as_scalar_tensor(lr, params), float(momentum),
            float(weight_decay), float(dampening), bool(nesterov),
            bool(first_step),
        )
        return
    # CPU oracle: upcast grads, run the reference path, recast shadow
    sgd_step_(
        params, grads_bf16.float(), momentum_buf, lr,
        momentum=momentum, weight_decay=weight_decay, dampening=dampening,
        nesterov=nesterov, first_step=first_step,
    )
    shadow.copy_(params)


def gather_multi_(ptr_table: torch.Tensor, offsets: torch.Tensor,
                  out: torch.Tensor) -> None:
    """Gather scattered grad tensors (device pointer table) into the
    flat buffer in one launch (steal-mode grads)."""
    _ext_for(out).gather_multi_(ptr_table, offsets, out)
