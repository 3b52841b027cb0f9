"""Thin communication layer over torch.distributed.

Maps the reference's MPI call-site inventory onto collectives:
  weight push  (per-layer fp64 MPI Bcast)  -> one fp32 broadcast of the flat
                                              parameter buffer
  grad pull    (pickled isend/irecv per layer x worker, tags 88+l)
                                           -> one gather of the fixed-layout
                                              wire bucket into a stacked
                                              (W, words) device tensor
  raw-sgd path                             -> one reduce (RCCL-native sum)
  step handshake (tag 10)                  -> implicit: collectives are the
                                              synchronization; step/lr are
                                              deterministic on every rank.
"""

from __future__ import annotations

import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist


class Comm:
    """Process-group wrapper; also usable un-initialized as a 1-process
    no-op comm (the N=1 self-PS mode and unit tests)."""

    def __init__(self, backend: Optional[str] = None, device: Optional[torch.device] = None):
        self.world = int(os.environ.get("WORLD_SIZE", "1"))
        self.rank = int(os.environ.get("RANK", "0"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", str(self.rank)))
        if device is not None:
            self.device = device
        elif torch.cuda.is_available():
            self.device = torch.device("cuda", self.local_rank % torch.cuda.device_count())
        else:
            self.device = torch.device("cpu")
        self.backend = backend or ("nccl" if self.device.type == "cuda" else "gloo")
        self._initialized = False
        if self.world > 1:
            if self.device.type == "cuda":
                torch.cuda.set_device(self.device)
            if not dist.is_initialized():
                os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
                os.environ.setdefault("MASTER_PORT", "29517")
                dist.init_process_group(
                    backend=self.backend,
                    rank=self.rank,
                    world_size=self.world,
                    timeout=datetime.timedelta(seconds=300),
                )
            self._initialized = True
    # -- collectives -----------------------------------------------------
    def broadcast(self, t: torch.Tensor, src: int = 0) -> None:
        if self._initialized:
            dist.broadcast(t, src=src)

    def reduce_sum(self, t: torch.Tensor, dst: int = 0) -> None:
        if self._initialized:
            dist.reduce(t, dst=dst, op=dist.ReduceOp.SUM)

    def all_reduce_sum(self, t: torch.Tensor) -> None:
        if self._initialized:
            dist.all_reduce(t, op=dist.ReduceOp.SUM)

    def gather(self, send: torch.Tensor, out_stacked: Optional[torch.Tensor], dst: int = 0) -> None:
        """Gather each rank's ``send`` (1-D) into rows of ``out_stacked``
        ((world, numel), significant on dst only)."""
        if not self._initialized:
            if out_stacked is not None:
                out_stacked[0].copy_(send)
            return
        # both gloo (CPU tests) and NCCL/RCCL implement gather; RCCL runs it
        # as grouped point-to-point sends into rank 0 — exactly the PS
        # topology's 7-inbound-xGMI-links pattern
        glist = (
            [out_stacked[i] for i in range(self.world)]
            if self.rank == dst
            else None
        )
        dist.gather(send, glist, dst=dst)

    def gather_arrival(
        self,
        send: torch.Tensor,
        out_stacked: Optional[torch.Tensor],
        dst: int = 0,
        on_arrival=None,
    ):
        """Point-to-point gather with ARRIVAL-ORDER notification — the
        reference's `MPI.Request.waitany` drain loop
        (sync_replicas_master_nn.py:198-215) mapped onto isend/irecv:
        the PS calls ``on_arrival(rank)`` as each worker's bucket lands
        (its own row first), so decode overlaps the remaining receives.
        Returns the arrival order on dst, else None."""
        if not self._initialized:
            if out_stacked is not None:
                out_stacked[0].copy_(send)
                if on_arrival is not None:
                    on_arrival(0)
            return [0] if out_stacked is not None else None
        if self.rank == dst:
            out_stacked[dst].copy_(send)
            reqs = {
                w: dist.irecv(out_stacked[w], src=w)
                for w in range(self.world)
                if w != dst
            }
            order = [dst]
            if on_arrival is not None:
                on_arrival(dst)
            pending = dict(reqs)
            while pending:
                done = [w for w, r in pending.items() if r.is_completed()]
                if not done:
                    # block on the longest-outstanding request
                    w = next(iter(pending))
                    pending.pop(w).wait()
                    done = [w]
                    done += [v for v, r in pending.items() if r.is_completed()]
                for w in done:
                    pending.pop(w, None)
                    order.append(w)
                    if on_arrival is not None:
                        on_arrival(w)
            return order
        dist.isend(send, dst=dst).wait()
        return None

    def barrier(self) -> None:
        if self._initialized:
            if self.backend == "nccl":
                dist.barrier(device_ids=[self.device.index])
            else:
                dist.barrier()

    def close(self) -> None:
        if self._initialized and dist.is_initialized():
            dist.destroy_process_group()
            self._initialized = False
