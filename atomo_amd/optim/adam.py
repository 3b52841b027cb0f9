"""Adam with externally-fed gradients over the flat parameter buffer
(reference math: optim/adam.py:37-93, incl. amsgrad option)."""

from __future__ import annotations

import math

import torch


class ExternalAdam:
    def __init__(
        self,
        flat_params: torch.Tensor,
        lr: float = 1e-3,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
        amsgrad: bool = False,
    ):
        assert flat_params.dim() == 1
        self.p = flat_params
        self.lr = float(lr)
        self.betas = betas
        self.eps = eps
        self.weight_decay = float(weight_decay)
        self.amsgrad = amsgrad
        self.t = 0
        self.exp_avg = torch.zeros_like(flat_params)
        self.exp_avg_sq = torch.zeros_like(flat_params)
        self.max_exp_avg_sq = torch.zeros_like(flat_params) if amsgrad else None

    @torch.no_grad()
    def step(self, flat_grad: torch.Tensor, grad_scale: float = 1.0) -> None:
        self.t += 1
        b1, b2 = self.betas
        if self.p.is_cuda:
            from ..ops import ext

            ext().fused_adam(
                self.p, flat_grad, self.exp_avg, self.exp_avg_sq,
                self.max_exp_avg_sq
                if self.max_exp_avg_sq is not None
                else torch.empty(0, device=self.p.device),
                self.lr, b1, b2, self.eps, self.weight_decay,
                1 - b1 ** self.t, 1 - b2 ** self.t, grad_scale,
            )
            return
        g = flat_grad
        if grad_scale != 1.0:
            g = g * grad_scale
        if self.weight_decay != 0.0:
            g = g.add(self.p, alpha=self.weight_decay)
        self.exp_avg.mul_(b1).add_(g, alpha=1 - b1)
        self.exp_avg_sq.mul_(b2).addcmul_(g, g, value=1 - b2)
        bias1 = 1 - b1 ** self.t
        bias2 = 1 - b2 ** self.t
        if self.amsgrad:
            torch.maximum(self.max_exp_avg_sq, self.exp_avg_sq, out=self.max_exp_avg_sq)
            denom = (self.max_exp_avg_sq / bias2).sqrt_().add_(self.eps)
        else:
            denom = (self.exp_avg_sq / bias2).sqrt_().add_(self.eps)
        step_size = self.lr / bias1
        self.p.addcdiv_(self.exp_avg, denom, value=-step_size)

    def state_dict(self):
        return {
            "lr": self.lr, "betas": self.betas, "eps": self.eps,
            "weight_decay": self.weight_decay, "amsgrad": self.amsgrad, "t": self.t,
            "exp_avg": self.exp_avg.clone(), "exp_avg_sq": self.exp_avg_sq.clone(),
            "max_exp_avg_sq": None if self.max_exp_avg_sq is None else self.max_exp_avg_sq.clone(),
        }

    def load_state_dict(self, sd):
        for k in ("lr", "betas", "eps", "weight_decay", "amsgrad", "t"):
            setattr(self, k, sd[k])
        # in place where shapes match (stable storage for any held refs)
        for name in ("exp_avg", "exp_avg_sq"):
            cur, new = getattr(self, name), sd[name].to(self.p.device)
            if cur is not None and cur.shape == new.shape:
                cur.copy_(new)
            else:
                setattr(self, name, new)
        if sd["max_exp_avg_sq"] is not None:
            new = sd["max_exp_avg_sq"].to(self.p.device)
            cur = getattr(self, "max_exp_avg_sq", None)
            if cur is not None and cur.shape == new.shape:
                cur.copy_(new)
            else:
                self.max_exp_avg_sq = new
