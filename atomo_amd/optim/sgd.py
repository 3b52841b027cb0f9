"""SGD with externally-fed gradients, fused over the flat parameter buffer.

Math identical to reference optim/sgd.py:57-89 (momentum buffer
``buf = mu*buf + d_p``, optional nesterov / weight decay) but the gradient is
one flat fp32 tensor (the PS aggregation buffer) and the whole update is a
single fused HIP kernel on MI355X (ops/csrc/atomo_kernels.hip) — the
reference loops per-parameter through numpy (SURVEY §2.10 row "SGD apply").
"""

from __future__ import annotations

import torch


class ExternalSGD:
    def __init__(
        self,
        flat_params: torch.Tensor,
        lr: float,
        momentum: float = 0.9,
        weight_decay: float = 0.0,
        nesterov: bool = False,
        dampening: float = 0.0,
    ):
        assert flat_params.dim() == 1
        self.p = flat_params
        self.lr = float(lr)
        self.momentum = float(momentum)
        self.weight_decay = float(weight_decay)
        self.nesterov = bool(nesterov)
        self.dampening = float(dampening)
        self.buf = (
            torch.zeros_like(flat_params) if self.momentum != 0.0 else None
        )

    @torch.no_grad()
    def step(self, flat_grad: torch.Tensor) -> None:
        if self.p.is_cuda:
            from ..ops import optim_ops

            optim_ops.fused_sgd(
                self.p,
                flat_grad,
                self.buf,
                lr=self.lr,
                momentum=self.momentum,
                weight_decay=self.weight_decay,
                nesterov=self.nesterov,
                dampening=self.dampening,
            )
            return
        d_p = flat_grad
        if self.weight_decay != 0.0:
            d_p = d_p.add(self.p, alpha=self.weight_decay)
        if self.momentum != 0.0:
            self.buf.mul_(self.momentum).add_(d_p, alpha=1.0 - self.dampening)
            d_p = d_p.add(self.buf, alpha=self.momentum) if self.nesterov else self.buf
        self.p.add_(d_p, alpha=-self.lr)

    def state_dict(self):
        return {
            "lr": self.lr,
            "momentum": self.momentum,
            "weight_decay": self.weight_decay,
            "nesterov": self.nesterov,
            "dampening": self.dampening,
            "buf": None if self.buf is None else self.buf.clone(),
        }

    def load_state_dict(self, sd):
        self.lr = sd["lr"]
        self.momentum = sd["momentum"]
        self.weight_decay = sd["weight_decay"]
        self.nesterov = sd["nesterov"]
        self.dampening = sd["dampening"]
        if sd["buf"] is not None:
            if self.buf is not None and self.buf.shape == sd["buf"].shape:
                # in place: a captured hipGraph holds this buffer's pointer
                self.buf.copy_(sd["buf"].to(self.p.device))
            else:
                self.buf = sd["buf"].to(self.p.device)
