"""Parameter-server data-parallel runtime over torch.distributed.

Reference topology: rank 0 is the parameter server, ranks
1..N-1 are workers; per step the PS pushes weights and gathers per-layer
coded gradients.  Here the transport is RCCL over xGMI (backend "nccl" on
ROCm; "gloo" for CPU tests): ONE bucketed broadcast of the flat fp32
parameter buffer down, ONE gather of each rank's fixed-layout wire bucket up
(or a single RCCL reduce for the raw ``sgd`` codec).  The PS is colocated
with a worker on rank 0 by default — on an 8-GPU MI355X node a dedicated
decode-only PS would idle 2.5 PF of matrix throughput; ``--dedicated-ps``
restores the reference's exact topology.
"""

from .comm import Comm
from .trainer import PSTrainer

__all__ = ["Comm", "PSTrainer"]
