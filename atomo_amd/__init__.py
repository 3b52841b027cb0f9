"""atomo_amd — MI355X-native atomic-sparsification data-parallel training engine.

A from-scratch rebuild of the capabilities of hwang595/ATOMO (NeurIPS 2018)
designed for AMD Instinct MI355X (gfx950, CDNA4):

* PyTorch-ROCm for autograd and the model zoo,
* hand-written HIP/CDNA4 kernels for the gradient codecs (SVD atomic
  sparsification, QSGD quantization) and the fused optimizer apply,
* RCCL over xGMI (torch.distributed "nccl" backend) for the rank-0
  parameter-server topology: one bucketed weight broadcast down, one
  fixed-layout coded-gradient gather up, per step.

Reference layer map: /root/reference/src.  This package is a
new design, not a port: weights live in one flat fp32 buffer (one RCCL
broadcast, one fused optimizer kernel), coded gradients travel in fixed-layout
device buckets (no pickle, no host hop), and the PS is colocated with a worker
rank by default so all N GPUs compute.
"""

__version__ = "0.1.0"

from . import codings, models, optim  # noqa: F401
