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
            view = self.flat.narrow(0, offset, n).view_as(t)
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
    """

    def __init__(self, module: torch.nn.Module, flatten_grads: bool = False):
        params = [p for p in module.parameters() if p.requires_grad]
        assert len(params) > 0, "module has no trainable parameters"
        dtypes = {p.dtype for p in params}
        assert len(dtypes) == 1, (
            f"FlatParams supports a single param dtype, got {dtypes}; "
            "use one FlatParams per dtype"
        )
        self.params = params
        self._buf = FlatBuffer([p.data for p in params])
        # re-point parameter storages at the flat views
        for p, v in zip(params, self._buf.views):
            p.data = v
        self.flat: torch.Tensor = self._buf.flat

        self.flat_grad: Optional[torch.Tensor] = None
        if flatten_grads:
            self.flat_grad = torch.zeros_like(self.flat)
            offset = 0
            for p in params:
                n = p.numel()
                p.grad = self.flat_grad.narrow(0, offset, n).view_as(p)
                offset += n

    def numel(self) -> int:
        return self.flat.numel()

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
            g = self.flat_grad.narrow(0, offset, n).view_as(p)
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
        self.rewire_grads()
