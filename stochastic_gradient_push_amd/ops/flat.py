"""Persistent flat parameter/gradient buffers.

The reference flattens its ~161 ResNet-50 parameter tensors with a fresh
``torch.cat`` every gossip step and scatters results back one tensor at a
time (reference gossip/utils/helpers.py:21-57, gossip/distributed.py:436-455).
On MI355X (288 GB HBM3E) we instead allocate the flat buffer once and
re-point every parameter at a view of it, so:

* the gossip message IS the parameter buffer (zero-copy pack),
* push-sum bias/de-bias/residual ops are single kernel launches over one
  contiguous ~100 MB range instead of ~161 launches,
* gradients optionally get the same treatment, enabling a one-launch fused
  SGD step for the whole model.
"""

from typing import List, Optional

import torch


def _format_view(flat: torch.Tensor, offset: int,
                 like: torch.Tensor) -> torch.Tensor:
    """A view of ``flat[offset:offset+numel]`` shaped like ``like`` and
    PRESERVING its memory format: a channels_last 4-D tensor gets a
    channels_last-strided view (elements stored NHWC inside the flat
    buffer), so re-pointing conv weights at the flat buffer never
    silently converts them to NCHW (MIOpen and the hand-written NHWC
    kernels both want channels_last weights)."""
    n = like.numel()
    sec = flat.narrow(0, offset, n)
    if like.dim() == 4 and like.is_contiguous(
        memory_format=torch.channels_last
    ) and not like.is_contiguous():
        N, C, H, W = like.shape
        return sec.view(N, H, W, C).permute(0, 3, 1, 2)
    return sec.view_as(like)


class FlatBuffer:
    """Flatten ``tensors`` (same dtype/device) into one contiguous buffer.

    After construction each original tensor's storage is replaced by a view
    into ``self.flat`` (classic flat-parameter trick; tensors remain leaf
    autograd tensors when they are ``nn.Parameter`` data).
    """

    def __init__(self, tensors: List[torch.Tensor]):
        assert len(tensors) > 0
        dtype, device = tensors[0].dtype, tensors[0].device
        assert all(t.dtype == dtype and t.device == device for t in tensors)
        self.shapes = [t.shape for t in tensors]
        self.numels = [t.numel() for t in tensors]
        total = sum(self.numels)
        self.flat = torch.empty(total, dtype=dtype, device=device)
        offset = 0
        self.views: List[torch.Tensor] = []
        for t, n in zip(tensors, self.numels):
            view = _format_view(self.flat, offset, t)
            view.copy_(t.detach())
            self.views.append(view)
            offset += n

    @property
    def dtype(self):
        return self.flat.dtype

    @property
    def device(self):
        return self.flat.device

    def numel(self) -> int:
        return self.flat.numel()

    def clone_flat(self) -> torch.Tensor:
        return self.flat.clone()


class FlatParams:
    """Flat view over a module's trainable parameters (single dtype).

    ``flatten_grads=True`` additionally pre-allocates a flat gradient
    buffer and points every ``p.grad`` at a view of it, so autograd
    accumulates directly into the contiguous buffer and the optimizer can
    run one fused kernel over the whole model.

    ``working_dtype=torch.bfloat16`` enables the mixed-precision master-
    weight layout: the flat buffer stays the fp32 MASTER (the optimizer
    and the push-sum gossip operate on it unchanged), but the matrix-
    shaped parameters (``ndim >= 2`` — convs and linears; BN scale/shift
    stay fp32 for the fused BN kernels) are re-pointed at a bf16 SHADOW
    of the master's leading section.  The model then computes directly
    with bf16 weights — removing every per-layer autocast weight-cast
    kernel — and their gradients arrive in a bf16 flat gradient buffer
    that the fused SGD consumes directly (ops.sgd_step_ bf16-grad
    variant refreshes the shadow in the same pass).  Parameter ORDER in
    the flat buffer is repartitioned (cast set first), identically on
    every rank, so gossip stays coherent.
    """

    def __init__(self, module: torch.nn.Module, flatten_grads: bool = False,
                 working_dtype: Optional[torch.dtype] = None):
        params = [p for p in module.parameters() if p.requires_grad]
        assert len(params) > 0, "module has no trainable parameters"
        dtypes = {p.dtype for p in params}
        assert len(dtypes) == 1, (
            f"FlatParams supports a single param dtype, got {dtypes}; "
            "use one FlatParams per dtype"
        )
        self.working_dtype = working_dtype
        if working_dtype is not None:
            assert working_dtype == torch.bfloat16
            assert flatten_grads, "working_dtype requires flatten_grads"
            cast = [p for p in params if p.ndim >= 2]
            keep = [p for p in params if p.ndim < 2]
            params = cast + keep
            self.n_cast = sum(p.numel() for p in cast)
        else:
            self.n_cast = 0
        self.params = params
        self._buf = FlatBuffer([p.data.float() if working_dtype is not None
                                else p.data for p in params])
        self.flat: torch.Tensor = self._buf.flat

        self.shadow: Optional[torch.Tensor] = None
        if working_dtype is not None:
            self.shadow = torch.empty(
                self.n_cast, dtype=working_dtype, device=self.flat.device
            )
            self.shadow.copy_(self.flat.narrow(0, 0, self.n_cast))
        # re-point parameter storages: cast set at shadow views, the
        # rest at fp32 master views (format-preserving, like the master
        # views themselves)
        offset = 0
        for p, v in zip(params, self._buf.views):
            if self.shadow is not None and offset < self.n_cast:
                p.data = _format_view(self.shadow, offset, p.data)
            else:
                p.data = v
            offset += p.numel()

        self.flat_grad: Optional[torch.Tensor] = None
        self.flat_grad_w: Optional[torch.Tensor] = None
        if flatten_grads:
            if self.shadow is not None:
                # bf16 grads for the cast set, fp32 for the rest
                self.flat_grad_w = torch.zeros(
                    self.n_cast, dtype=working_dtype,
                    device=self.flat.device,
                )
                self.flat_grad = torch.zeros(
                    self.flat.numel() - self.n_cast,
                    dtype=self.flat.dtype, device=self.flat.device,
                )
            else:
                self.flat_grad = torch.zeros_like(self.flat)
            self.rewire_grads()

    def numel(self) -> int:
        return self.flat.numel()

    # -- steal-mode gradients ------------------------------------------------
    # With p.grad = None, autograd ASSIGNS each produced gradient tensor
    # (no accumulate-add kernel per parameter).  gather_grads() then
    # concatenates all of them into the flat buffers with ONE kernel per
    # dtype group — replacing ~161 tiny adds per step.  Valid for the
    # single-backward-per-step regime (no micro-batch accumulation).

    def enable_steal_mode(self) -> None:
        assert self.flat_grad is not None
        if getattr(self, "steal_mode", False):
            return
        self.steal_mode = True
        dev = self.flat.device
        groups = []
        if self.shadow is not None:
            cast = [p for p in self.params if p.ndim >= 2]
            keep = [p for p in self.params if p.ndim < 2]
            groups.append((cast, self.flat_grad_w))
            groups.append((keep, self.flat_grad))
        else:
            groups.append((self.params, self.flat_grad))
        self._steal_groups = []
        for params, out in groups:
            if not params:
                continue
            offs = [0]
            for p in params:
                offs.append(offs[-1] + p.numel())
            offsets = torch.tensor(offs, dtype=torch.int64, device=dev)
            host = torch.zeros(len(params), dtype=torch.int64)
            if dev.type == "cuda":
                host = host.pin_memory()
            table = torch.zeros(len(params), dtype=torch.int64, device=dev)
            self._steal_groups.append(
                {"params": params, "out": out, "offsets": offsets,
                 "host": host, "table": table, "ptrs": None}
            )
        for p in self.params:
            p.grad = None

    def gather_grads(self) -> None:
        """One fused gather of every scattered autograd grad into the
        flat buffers (call after backward, before the optimizer)."""
        assert getattr(self, "steal_mode", False)
        from . import gather_multi_

        for grp in self._steal_groups:
            params, out = grp["params"], grp["out"]
            fixups = set()
            ptrs = []
            for p in params:
                g = p.grad
                if g is None:
                    ptrs.append(0)
                elif g.stride() == p.data.stride() or g.numel() <= 1:
                    ptrs.append(g.data_ptr())
                else:
                    # unexpected layout: zero-fill + per-param copy
                    ptrs.append(0)
                    fixups.add(id(p))
            if ptrs != grp["ptrs"]:
                capturing = out.is_cuda and \
                    torch.cuda.is_current_stream_capturing()
                if grp.get("captured") and not capturing:
                    # a hipGraph replay re-executes the captured H2D
                    # table copy, re-reading THIS host buffer — an
                    # eager step must never mutate it, so it gets a
                    # fresh host+device table pair of its own
                    host = torch.zeros(len(params), dtype=torch.int64)
                    if out.is_cuda:
                        host = host.pin_memory()
                    grp["host"] = host
                    grp["table"] = torch.zeros(
                        len(params), dtype=torch.int64, device=out.device
                    )
                grp["captured"] = capturing
                grp["host"].copy_(torch.tensor(ptrs, dtype=torch.int64))
                grp["table"].copy_(grp["host"], non_blocking=True)
                grp["ptrs"] = ptrs
            if out.is_cuda:
                gather_multi_(grp["table"], grp["offsets"], out)
            else:
                offs = 0
                for p in params:
                    n = p.numel()
                    if p.grad is None:
                        out.narrow(0, offs, n).zero_()
                    else:
                        _format_view(out, offs, p.data).copy_(
                            p.grad.detach()
                        )
                    offs += n
            if fixups:
                offs = 0
                for p in params:
                    n = p.numel()
                    if id(p) in fixups:
                        _format_view(out, offs, p.data).copy_(
                            p.grad.detach()
                        )
                    offs += n

    def release_grads(self) -> None:
        """Drop the scattered grad tensors so the next backward assigns
        fresh ones (steal-mode zero_grad — no fill kernel needed)."""
        for p in self.params:
            p.grad = None

    def grads_wired(self) -> bool:
        """True when the first parameter's ``.grad`` still aliases the
        flat gradient buffer (the cheap per-step wiring check)."""
        if self.flat_grad is None:
            return True
        p0 = self.params[0]
        ref = (self.flat_grad_w
               if self.flat_grad_w is not None and self.n_cast
               else self.flat_grad)
        return p0.grad is not None and p0.grad.data_ptr() == ref.data_ptr()

    def sync_shadow(self) -> None:
        """Refresh the bf16 working weights from the fp32 master (call
        after anything other than the fused SGD mutates the master —
        gossip merges, load_state_dict, bias/de-bias rescales)."""
        if self.shadow is None:
            return
        from . import cast_shadow_

        cast_shadow_(self.flat.narrow(0, 0, self.n_cast), self.shadow)

    def rewire_grads(self) -> None:
        """Re-point ``p.grad`` at the flat grad views.

        Call after something replaced them — typically
        ``optimizer.zero_grad(set_to_none=True)``.  Semantics are
        preserved exactly: a param whose grad was ``None`` gets a *zeroed*
        view (autograd would have allocated a zero-initialized fresh
        tensor); a param holding a foreign grad tensor gets its values
        copied into the view (accumulation history kept).
        """
        assert self.flat_grad is not None
        offset = 0
        for p in self.params:
            n = p.numel()
            if self.shadow is not None and offset < self.n_cast:
                g = _format_view(self.flat_grad_w, offset, p.data)
            else:
                base = offset - self.n_cast
                g = _format_view(self.flat_grad, base, p.data)
            if p.grad is None:
                g.detach().zero_()
                p.grad = g
            elif p.grad.data_ptr() != g.data_ptr():
                g.detach().copy_(p.grad.detach())
                p.grad = g
            offset += n

    def zero_grad(self) -> None:
        assert self.flat_grad is not None
        self.flat_grad.zero_()
        if self.flat_grad_w is not None:
            self.flat_grad_w.zero_()
        self.rewire_grads()
