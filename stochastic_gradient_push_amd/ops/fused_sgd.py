"""FusedSGD — one-kernel momentum SGD over a flat parameter buffer.

The reference relies on ``torch.optim.SGD`` looping over ~161 ResNet-50
tensors per step (and the AD-PSGD gossip process keeps its own SGD,
reference ad_psgd.py:261-266).  With parameters and gradients living in
flat buffers (:class:`~stochastic_gradient_push_amd.ops.flat.FlatParams`)
the whole update is a single gfx950 kernel launch
(:func:`~stochastic_gradient_push_amd.ops.sgd_step_`): 3 reads + 2 writes
of ~100 MB at HBM speed instead of hundreds of launches.

Drop-in for the common SGD surface (param_groups with 'lr', zero_grad,
step, state_dict/load_state_dict).
"""



import torch

from . import sgd_step_
from .flat import FlatParams
from ..utils.roctx import trace as _roctx


class FusedSGD:
    def __init__(
        self,
        flatp: FlatParams,
        lr: float,
        momentum: float = 0.0,
        weight_decay: float = 0.0,
        dampening: float = 0.0,
        nesterov: bool = False,
        steal_grads: bool = False,
    ):
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError(
                "Nesterov momentum requires momentum > 0 and dampening == 0"
            )
        assert flatp.flat_grad is not None, (
            "FusedSGD requires FlatParams(flatten_grads=True)"
        )
        self.flatp = flatp
        self._mixed = flatp.shadow is not None
        self._steal = steal_grads
        if steal_grads:
            # autograd assigns fresh grad tensors (no per-param
            # accumulate adds); step() gathers them in one kernel
            flatp.enable_steal_mode()
        if momentum != 0.0:
            self.momentum_buf = torch.zeros_like(flatp.flat)
        elif self._mixed:
            self.momentum_buf = torch.zeros_like(flatp.flat)
        else:
            self.momentum_buf = flatp.flat_grad
        self._first_step = True
        # device-resident lr: the step kernel reads it by pointer, so a
        # hipGraph-captured step sees live schedule updates (sync_lr())
        self._lr_dev = torch.tensor(
            [float(lr)],
            device=flatp.flat.device, dtype=torch.float32,
        ) if flatp.flat.is_cuda else None
        self._lr_host = float(lr)
        # torch-optim-style param_groups so LR schedules written against
        # torch.optim keep working
        self.param_groups = [
            {
                "lr": lr,
                "momentum": momentum,
                "weight_decay": weight_decay,
                "dampening": dampening,
                "nesterov": nesterov,
                "params": list(flatp.params),
            }
        ]

    def zero_grad(self, set_to_none: bool = False):
        if self._steal:
            # drop the scattered tensors; nothing to zero (the gather
            # overwrites the whole flat buffer next step)
            self.flatp.release_grads()
            return
        # flat buffers are zeroed, never detached — views stay wired
        self.flatp.zero_grad()

    def sync_lr(self):
        """Copy param_groups[0]['lr'] into the device scalar.  Called
        automatically by step(); call it manually after LR-schedule
        changes when the step itself is inside a captured hipGraph."""
        lr = float(self.param_groups[0]["lr"])
        if self._lr_dev is not None and lr != self._lr_host:
            if torch.cuda.is_current_stream_capturing():
                # never bake the fill_ into a captured hipGraph — a
                # captured constant would silently freeze the schedule
                # on every replay; the post-capture sync_lr() call (or
                # the next eager step) lands the update instead
                return
            self._lr_dev.fill_(lr)
            self._lr_host = lr

    @torch.no_grad()
    def step(self, closure=None):
        g = self.param_groups[0]
        if self._steal:
            self.flatp.gather_grads()
        # autograd may have detached grads (e.g. someone else's zero_grad)
        elif not self.flatp.grads_wired():
            self.flatp.rewire_grads()
        self.sync_lr()
        lr = self._lr_dev if self._lr_dev is not None else g["lr"]
        with _roctx("sgp:fused_sgd_step"):
            if self._mixed:
                # cast-set section: fp32 master <- bf16 grads, bf16
                # working-weight shadow refreshed in the same kernel
                from . import sgd_step_bf16gs_

                n_cast = self.flatp.n_cast
                sgd_step_bf16gs_(
                    self.flatp.flat.narrow(0, 0, n_cast),
                    self.flatp.flat_grad_w,
                    self.momentum_buf.narrow(0, 0, n_cast),
                    self.flatp.shadow,
                    lr=lr,
                    momentum=g["momentum"],
                    weight_decay=g["weight_decay"],
                    dampening=g["dampening"],
                    nesterov=g["nesterov"],
                    first_step=self._first_step,
                )
                rest = self.flatp.flat.numel() - n_cast
                if rest:
                    sgd_step_(
                        self.flatp.flat.narrow(0, n_cast, rest),
                        self.flatp.flat_grad,
                        self.momentum_buf.narrow(0, n_cast, rest),
                        lr=lr,
                        momentum=g["momentum"],
                        weight_decay=g["weight_decay"],
                        dampening=g["dampening"],
                        nesterov=g["nesterov"],
                        first_step=self._first_step,
                    )
            else:
                sgd_step_(
                    self.flatp.flat,
                    self.flatp.flat_grad,
                    self.momentum_buf,
                    lr=lr,
                    momentum=g["momentum"],
                    weight_decay=g["weight_decay"],
                    dampening=g["dampening"],
                    nesterov=g["nesterov"],
                    first_step=self._first_step,
                )
        self._first_step = False

    def state_dict(self):
        return {
            "momentum_buf": self.momentum_buf,
            "first_step": self._first_step,
            "hyper": {
                k: v for k, v in self.param_groups[0].items() if k != "params"
            },
        }

    def load_state_dict(self, sd):
        self.momentum_buf.copy_(sd["momentum_buf"])
        self._first_step = sd["first_step"]
        self.param_groups[0].update(sd["hyper"])
