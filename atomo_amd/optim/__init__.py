"""Externally-fed-gradient optimizers (reference: src/optim/sgd.py,
src/optim/adam.py — gradients come from the PS aggregation buffer, not
p.grad).  The reference stores an Adam import it never wires to a flag
(sync_replicas_master_nn.py:21); here ``--optimizer {sgd,adam}`` selects it.
"""

from .sgd import ExternalSGD
from .adam import ExternalAdam


def make_optimizer(name, flat_params, lr, momentum=0.9, weight_decay=0.0, **kw):
    if name == "sgd":
        return ExternalSGD(flat_params, lr=lr, momentum=momentum,
                           weight_decay=weight_decay, **kw)
    if name == "adam":
        return ExternalAdam(flat_params, lr=lr, weight_decay=weight_decay)
    raise ValueError(f"unknown optimizer {name!r}")


__all__ = ["ExternalSGD", "ExternalAdam", "make_optimizer"]
