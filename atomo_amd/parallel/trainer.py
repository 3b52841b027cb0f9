"""Synchronous parameter-server trainer.

One object runs on every rank; role by rank (reference step machines:
SyncReplicasMaster_NN.train, sync_replicas_master_nn.py:173-234, and
DistributedWorker.train, distributed_worker.py:166-262).  Per global step:

  1. fetch   broadcast of the flat fp32 parameter buffer from rank 0
             (replaces per-layer fp64 Bcast + step handshake tags 10).
  2. comp    forward/backward on this rank's synthetic batch (skipped by a
             dedicated PS).
  3. encode  per-layer codec encode straight into the fixed wire bucket.
  4. comm    ONE gather of wire buckets to rank 0 (ONE reduce for the raw
             codec — RCCL sums on the wire; ``comm_type="P2P"`` switches to
             arrival-order isend/irecv with decode-as-arrives and optional
             first-K ``num_aggregate`` partial aggregation).
  5. decode  PS accumulates all workers' packets into the flat agg buffer
             (fused HIP kernels on GPU).
  6. apply   external-grad optimizer step with grad_scale = 1/num_workers,
             then lr shrinkage every ``shrink_freq`` steps (reference
             hardcodes 50, sync_replicas_master_nn.py:103-107).

Full-sync semantics are preserved: every rank participates in both
collectives every step, so the PS implicitly collects exactly
``num_workers`` gradients per step (reference counter semantics,
sync_replicas_master_nn.py:113,212-214).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn as nn

from ..codings import Codec
from ..models import build_model
from ..optim import make_optimizer
from ..utils import PhaseTimers, flatten_params
from .comm import Comm
from .wire import WireCodec


class PSTrainer:
    def __init__(
        self,
        model_name: str,
        codec: Codec,
        comm: Comm,
        lr: float = 0.01,
        momentum: float = 0.9,
        weight_decay: float = 0.0,
        optimizer: str = "sgd",
        lr_shrinkage: float = 0.95,
        shrink_freq: int = 50,
        num_classes: int = 10,
        in_channels: int = 3,
        dedicated_ps: bool = False,
        device: Optional[torch.device] = None,
        seed: Optional[int] = None,
        checkpoint_freq: int = 0,
        train_dir: str = "output/models/",
        use_graph: bool = False,
        overlap: bool = False,
        step_timeout: float = 0.0,
        defer_loss: bool = False,
        amp: bool = False,
        comm_type: str = "Bcast",
        num_aggregate: int = 0,
        wire_dtype: str = "fp32",
    ):
        self.comm = comm
        self.device = device or comm.device
        if self.device.type == "cuda":
            # GPU ranks: host work is tiny batched eigensolves + sampling —
            # hundreds of OpenMP threads (EPYC boxes) make a 6 ms LAPACK
            # call take 600 ms.  Cap them.
            torch.set_num_threads(min(8, os.cpu_count() or 8))
        if seed is not None:
            # rank-decorrelated: host-side compression draws (sample_svd,
            # QSGD dice) ride the global torch RNG — every worker needs its
            # own stream.  Model-init consistency is guaranteed separately
            # by the flat-param broadcast from rank 0 below.
            torch.manual_seed(seed + comm.rank * 1000003)
        self.model = build_model(model_name, num_classes, in_channels).to(self.device)
        self.flat, self.params = flatten_params(self.model)
        self.wc = WireCodec(codec, self.params, self.device, rank=comm.rank)
        self.codec = codec
        self.loss_fn = nn.CrossEntropyLoss()

        self.dedicated_ps = dedicated_ps and comm.world > 1
        # P2P comm mode: arrival-order gather + decode-as-arrives (the
        # reference's waitany drain, sync_replicas_master_nn.py:198-215);
        # num_aggregate > 0 aggregates only the first K arrivals (the
        # reference stores this flag but never implements it — here the
        # PS drops late gradients from the update, full-sync timing).
        self.p2p = comm_type.lower() in ("p2p", "isend") and comm.world > 1
        self.num_aggregate = int(num_aggregate or 0)
        self._last_contrib = 0
        self.is_master = comm.rank == 0
        self.is_worker = (not self.dedicated_ps) or comm.rank > 0
        self.num_workers = comm.world - 1 if self.dedicated_ps else comm.world

        # gradients live in one flat buffer (views as p.grad): zeroing is one
        # kernel, and for the raw codec the wire IS the grad buffer (layer
        # order and sizes coincide), so encode is free and RCCL reduces the
        # gradients in place.
        self.flat_grad = torch.zeros_like(self.flat)
        off = 0
        for p in self.params:
            n = p.numel()
            p.grad = self.flat_grad[off : off + n].view_as(p)
            off += n
        if self.wc.reducible:
            assert self.wc.total_words == self.flat.numel()
            self.wire = self.flat_grad
        else:
            self.wire = torch.zeros(
                self.wc.total_words, dtype=torch.float32, device=self.device
            )
        self.gather_buf = (
            torch.zeros(
                max(1, comm.world),
                self.wc.total_words,
                dtype=torch.float32,
                device=self.device,
            )
            if self.is_master and not self.wc.reducible and comm.world > 1
            else None
        )
        if self.p2p and self.wc.reducible:
            self.p2p = False  # raw codec always rides the RCCL reduce
        # depth-D pipelined receive buffers for TRUE partial aggregation
        # (the PS returns after K arrivals; late receives stay pending
        # across steps and are dropped as stale when they land)
        self.pp_buf = (
            torch.zeros(
                comm.PIPE_DEPTH,
                comm.world,
                self.wc.total_words,
                dtype=torch.float32,
                device=self.device,
            )
            if self.p2p and self.is_master
            else None
        )
        self.agg = (
            torch.zeros_like(self.flat)
            if self.is_master and not self.wc.reducible
            else None
        )
        # opt-in bf16 WIRE: halves the bytes of the weight push and of the
        # SVD factor packets over xGMI (justified by the
        # N=8 comm phase).  fp32 stays the default — and the master copy,
        # optimizer state and decode/apply all stay fp32.  Invalid for
        # QSGD: its wire words are packed integer bit patterns that a
        # dtype cast would destroy.
        self.wire_bf16 = str(wire_dtype).lower() in ("bf16", "bfloat16")
        if self.wire_bf16:
            from ..codings import QSGDCodec

            if isinstance(codec, QSGDCodec):
                raise ValueError(
                    "--wire-dtype bf16 is invalid for qsgd: its wire words "
                    "are packed integer bit patterns"
                )
            # raw codec: only the weight push narrows; the gradient
            # reduce stays fp32 (exact sums)
            self._bcast_bf = torch.empty_like(self.flat, dtype=torch.bfloat16)
            if comm.world > 1 and not self.wc.reducible:
                self._wire_bf = torch.empty(
                    self.wc.total_words, dtype=torch.bfloat16, device=self.device
                )
                self._gather_bf = (
                    torch.empty(
                        comm.world,
                        self.wc.total_words,
                        dtype=torch.bfloat16,
                        device=self.device,
                    )
                    if self.is_master
                    else None
                )
        self.opt = (
            make_optimizer(
                optimizer, self.flat, lr=lr, momentum=momentum, weight_decay=weight_decay
            )
            if self.is_master
            else None
        )
        self.lr = lr
        self.lr_shrinkage = lr_shrinkage
        self.shrink_freq = shrink_freq
        self.step_num = 0
        self.checkpoint_freq = checkpoint_freq
        self.train_dir = train_dir
        self.timers = PhaseTimers()
        self.last_loss = float("nan")
        # defer_loss: keep the loss on device (no per-step host sync); read
        # it via current_loss() when logging.
        self.defer_loss = bool(defer_loss)
        self._loss_tensor = None
        # opt-in bf16 autocast for forward/backward (gradients and the whole
        # codec/optimizer path stay fp32; NOT used by the default benchmark,
        # whose dtype matches the reference's fp32)
        self.amp = bool(amp) and self.device.type == "cuda"
        self.watchdog = None
        if step_timeout and step_timeout > 0:
            from ..utils.watchdog import StepWatchdog

            self.watchdog = StepWatchdog(
                step_timeout, action="abort", rank=comm.rank
            ).start()
        # hipGraph capture of forward+backward: the per-step launch storm of
        # a convnet (hundreds of kernels) replays as one graph.  Input/label
        # buffers are static; flat_grad.zero_() is captured inside, so
        # autograd accumulation stays correct.
        self.use_graph = bool(use_graph) and self.device.type == "cuda"
        self._graph = None
        self._static_x = None
        self._static_y = None
        self._static_loss = None
        # WHOLE-STEP graph (fwd/bwd + encode + decode + apply in ONE
        # replay): eligible when the step is a pure device-kernel pipeline
        # — single process, raw codec (wire aliases grads) or the fully-
        # device SVD path (device sampler seed rides pinned memory so each
        # replay draws fresh atoms).  Collectives are no-ops at world==1.
        self._wgraph = None
        self.graph_whole = False
        # device-resident lr for the fused apply: a captured pinned copy
        # re-reads it per replay, so lr shrinkage never forces a recapture
        if self.device.type == "cuda" and self.is_master:
            self._lr_host = torch.zeros(1, pin_memory=True)
            self._lr_dev = torch.zeros(1, device=self.device)
        else:
            self._lr_host = self._lr_dev = None
        if self.use_graph and self.amp:
            # the captured bodies run fp32; don't silently drop autocast
            self.use_graph = False
        if (
            self.use_graph
            and self.is_master
            and type(self.opt).__name__ == "ExternalAdam"
        ):
            # Adam's per-step bias corrections are scalar kernel args — a
            # captured apply would freeze them at the capture step
            self.use_graph = False
        if self.use_graph and comm.world == 1 and not self.dedicated_ps:
            if self.wc.reducible:
                self.graph_whole = True
            elif self.wc._qsgd_tables is not None:
                self.graph_whole = True  # batched pack/unpack, device seed
            else:
                enc = self.wc._batched_encoder
                self.graph_whole = bool(
                    enc is not None
                    and enc.use_kernels
                    and not enc.exact_eigh  # hipSOLVER syevd: not capturable
                    and len(enc.kernel_set) == len(self.wc.specs)
                    and self.codec.generator is None
                )
        # SPLIT graphs at world>1 (collective-gather mode, colocated):
        # capture (zero+fwd/bwd+encode) and, on the PS, (decode+apply) as
        # two graphs with the eager RCCL gather/broadcast between replays
        # — recovers the whole-step graph's launch-latency win at N>1
        # without capturing collectives.
        self.graph_split = False
        if (
            self.use_graph
            and comm.world > 1
            and not self.dedicated_ps
            and not self.p2p
            and not self.wire_bf16
        ):
            if self.wc.reducible:
                self.graph_split = True
            elif self.wc._qsgd_tables is not None:
                self.graph_split = True
            else:
                enc = self.wc._batched_encoder
                self.graph_split = bool(
                    enc is not None
                    and enc.use_kernels
                    and not enc.exact_eigh  # hipSOLVER syevd: not capturable
                    and len(enc.kernel_set) == len(self.wc.specs)
                    and self.codec.generator is None
                )
        self._graphA = None
        self._graphB = None
        if self.use_graph and not self.graph_whole and not self.graph_split:
            # capture ineligible (host layers / pinned RNG / exact-eigh
            # oracle / amp / Adam): backward-hook overlap beats a
            # fwd/bwd-only graph for the svd encode path, so let it win
            self.use_graph = False
        self.overlap = (
            bool(overlap)
            and self.device.type == "cuda"
            and self.is_worker
            and not self.use_graph
        )
        self._side_stream = None
        if self.overlap:
            self._side_stream = torch.cuda.Stream()
            self.overlap = self.wc.setup_overlap(
                self._side_stream, self.params, self.wire
            )

        self.skip_count = 0  # straggler self-skips (partial mode, worker)
        if self.p2p and self.num_aggregate > 0 and self.is_master:
            self.comm.publish_step(0)
        # make every rank start from rank-0's init
        self.comm.broadcast(self.flat, src=0)

    # -----------------------------------------------------------------
    def train_step(self, x: torch.Tensor, y: torch.Tensor) -> float:
        t = self.timers
        if self.graph_whole:
            with t.phase("comp"):
                done = self._whole_step_graphed(x, y)
            if done:
                if self.wc.reducible or self.wc._qsgd_tables is not None:
                    # fixed-layout wires ship every word each step (the
                    # svd codec's variable packets are counted on device)
                    t.add_scalar("msg_bytes", 4.0 * self.wc.total_words)
                self._last_contrib = self.num_workers
                self._post_step()
                return self.last_loss
        if self.graph_split and self.step_num >= 1:
            # step 0 runs the standard eager path so every decode/apply
            # kernel is warm before graph B captures
            return self._train_step_split(x, y)
        with t.phase("fetch"):
            if self.p2p and self.num_aggregate > 0:
                # partial mode: per-worker pipelined weight push — a
                # broadcast collective would rendezvous with a straggler
                # every step and stall the PS (see Comm.gather_partial)
                if self.is_master:
                    self.comm.ps_push_weights(self.flat, self.step_num)
                else:
                    self.comm.recv_weights(self.flat, src=0)
            elif self.wire_bf16 and self.comm.world > 1:
                if self.is_master:
                    self._bcast_bf.copy_(self.flat)
                self.comm.broadcast(self._bcast_bf, src=0)
                if not self.is_master:
                    self.flat.copy_(self._bcast_bf)
            else:
                self.comm.broadcast(self.flat, src=0)

        skip_compute = False
        if self.p2p and self.num_aggregate > 0 and not self.is_master:
            # straggler self-skip (tag-77 kill redesigned, see Comm): a
            # worker >= D-2 steps behind ships a valid zero packet and
            # catches up instead of throttling the pipeline
            behind = self.comm.ps_step_behind(self.step_num)
            if behind >= self.comm.PIPE_DEPTH - 2:
                skip_compute = True
                self.skip_count += 1
                self.wire.zero_()

        if self.is_worker and not skip_compute:
            with t.phase("comp"):
                self.model.train()
                if self.overlap:
                    self.wc.arm_overlap()
                if self.use_graph and not self.graph_split:
                    # (split mode's eager step 0 must not capture the
                    # fwd/bwd-only graph it would never replay)
                    self._fwd_bwd_graphed(x, y)
                else:
                    self.flat_grad.zero_()
                    if self.amp:
                        with torch.autocast("cuda", dtype=torch.bfloat16):
                            out = self.model(x)
                            loss = self.loss_fn(out, y)
                    else:
                        out = self.model(x)
                        loss = self.loss_fn(out, y)
                    loss.backward()
                    if self.defer_loss:
                        self._loss_tensor = loss.detach()
                    else:
                        self.last_loss = float(loss.detach())
                if self.overlap:
                    torch.cuda.current_stream().wait_stream(self._side_stream)
            with t.phase("encode"):
                if self.wc.reducible:
                    used = self.wc.total_words  # wire aliases flat_grad
                else:
                    used = self.wc.encode_all(
                        self.wire,
                        flat_grad=self.flat_grad,
                        overlap_done=self.overlap,
                    )
                if used >= 0:
                    t.add_scalar("msg_bytes", 4.0 * used)
        elif self.wc.reducible and not self.is_worker:
            self.flat_grad.zero_()  # dedicated PS contributes zeros to the sum

        if self.p2p and not self.wc.reducible:
            with t.phase("comm"):
                # Arrival-order P2P gather (the reference's waitany drain)
                # with decode-as-arrives.  num_aggregate > 0 is TRUE
                # partial aggregation: the PS returns after the first K
                # contributions; a straggler's packet stays pending and
                # is dropped as stale when it lands.  The colocated PS's
                # own contribution is always in the first-K set — that is
                # physical (zero latency); which K-1 workers fill the
                # rest is genuine arrival order, not rank order (ADVICE
                # r1 fairness note: with a persistent straggler the
                # EXCLUDED set is whoever is actually late, not a fixed
                # rank).
                contrib = [0]
                if self.is_master:
                    self.agg.zero_()

                    def on_arr(w, sl):
                        self.wc.decode_all(
                            self.pp_buf[sl, w : w + 1], self.agg
                        )
                        contrib[0] += 1

                    self.comm.gather_partial(
                        self.wire,
                        self.pp_buf,
                        self.step_num,
                        dst=0,
                        target=self.num_aggregate,
                        on_arrival=on_arr,
                        self_counts=not self.dedicated_ps,
                    )
                else:
                    self.comm.gather_partial(
                        self.wire, None, self.step_num, dst=0,
                        target=self.num_aggregate,
                    )
                    contrib[0] = self.num_aggregate or self.num_workers
            if self.num_aggregate > 0 and self.is_master:
                self.comm.publish_step(self.step_num + 1)
            self._last_contrib = contrib[0]
            grad_flat = self.agg if self.is_master else None
        else:
            with t.phase("comm"):
                if self.wc.reducible:
                    self.comm.reduce_sum(self.wire, dst=0)
                elif self.comm.world > 1:
                    if self.wire_bf16:
                        self._wire_bf.copy_(self.wire)
                        self.comm.gather(self._wire_bf, self._gather_bf, dst=0)
                        if self.is_master:
                            self.gather_buf.copy_(self._gather_bf)
                    else:
                        self.comm.gather(self.wire, self.gather_buf, dst=0)

            if self.is_master:
                with t.phase("decode"):
                    if self.wc.reducible:
                        grad_flat = self.wire
                    else:
                        self.agg.zero_()
                        rows = (
                            self.gather_buf
                            if self.comm.world > 1
                            else self.wire.view(1, -1)
                        )
                        self.wc.decode_all(rows, self.agg)
                        grad_flat = self.agg
            self._last_contrib = self.num_workers

        if self.is_master:
            with t.phase("apply"):
                self.opt.lr = self.lr
                self._apply(grad_flat)

        self.step_num += 1
        if self.watchdog is not None:
            self.watchdog.step()
        if self.step_num % self.shrink_freq == 0:
            self.lr *= self.lr_shrinkage
        if (
            self.is_master
            and self.checkpoint_freq > 0
            and self.step_num % self.checkpoint_freq == 0
        ):
            self.save_checkpoint()
        return self.last_loss

    def current_loss(self) -> float:
        if self._loss_tensor is not None:
            self.last_loss = float(self._loss_tensor)
        return self.last_loss

    def _quarantine_graph(self, g) -> None:
        """A partially-captured CUDAGraph aborts the process in its
        destructor (HIPGeneratorImpl 'graph should be registered');
        reset it if possible and keep the object alive."""
        if g is None:
            return
        try:
            g.reset()
        except Exception:
            pass
        if not hasattr(self, "_dead_graphs"):
            self._dead_graphs = []
        self._dead_graphs.append(g)

    def _post_step(self) -> None:
        self.step_num += 1
        if self.watchdog is not None:
            self.watchdog.step()
        if self.step_num % self.shrink_freq == 0:
            self.lr *= self.lr_shrinkage
        if (
            self.is_master
            and self.checkpoint_freq > 0
            and self.step_num % self.checkpoint_freq == 0
        ):
            self.save_checkpoint()

    def _body_A(self) -> None:
        """Pre-comm half of a split-graph step: zero + fwd/bwd + encode
        on static buffers (no optimizer mutation — safe to warm-run)."""
        self.flat_grad.zero_()
        loss = self.loss_fn(self.model(self._static_x), self._static_y)
        loss.backward()
        self._static_loss = loss.detach()
        if not self.wc.reducible:
            self.wc.encode_all(self.wire, flat_grad=self.flat_grad)

    def _graphA_step(self, x: torch.Tensor, y: torch.Tensor) -> bool:
        if self._graphA is None:
            g = None
            try:
                self._static_x = x.clone()
                self._static_y = y.clone()
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    for _ in range(3):
                        self._body_A()
                torch.cuda.current_stream().wait_stream(side)
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g, capture_error_mode="thread_local"):
                    self._body_A()
                self._graphA = g
            except Exception as exc:
                print(
                    f"[atomo] split-graph A capture failed ({exc}); eager",
                    flush=True,
                )
                self.graph_split = False
                self._quarantine_graph(g)
                return False
        self._static_x.copy_(x)
        self._static_y.copy_(y)
        self.wc.advance_seeds()
        self._graphA.replay()
        if self.defer_loss:
            self._loss_tensor = self._static_loss
        else:
            self.last_loss = float(self._static_loss)
        return True

    def _graphB_step(self) -> bool:
        """PS-side decode+apply graph (capture needs kernels warmed by an
        eager step 0; lr rides device memory, so shrinkage never forces a
        recapture)."""
        if self._graphB is None:
            g = None
            try:
                self.opt.lr = self.lr
                self._lr_host[0] = self.lr
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g, capture_error_mode="thread_local"):
                    if self.wc.reducible:
                        self._apply(self.flat_grad)
                    else:
                        self.agg.zero_()
                        self.wc.decode_all(self.gather_buf, self.agg)
                        self._apply(self.agg)
                self._graphB = g
            except Exception as exc:
                print(
                    f"[atomo] split-graph B capture failed ({exc}); eager",
                    flush=True,
                )
                self._quarantine_graph(g)
                self._graphB = None
                self.graph_split = False
                return False
        self._lr_host[0] = self.lr  # replay re-reads the pinned lr
        self._graphB.replay()
        return True

    def _train_step_split(self, x: torch.Tensor, y: torch.Tensor) -> float:
        t = self.timers
        with t.phase("fetch"):
            self.comm.broadcast(self.flat, src=0)
        with t.phase("comp"):
            ok = self._graphA_step(x, y)
            if not ok:  # capture failed (graph_split now False): eager body
                self.flat_grad.zero_()
                loss = self.loss_fn(self.model(x), y)
                loss.backward()
                if self.defer_loss:
                    self._loss_tensor = loss.detach()
                else:
                    self.last_loss = float(loss.detach())
                if not self.wc.reducible:
                    self.wc.encode_all(self.wire, flat_grad=self.flat_grad)
        if self.wc.reducible or self.wc._qsgd_tables is not None:
            t.add_scalar("msg_bytes", 4.0 * self.wc.total_words)
        with t.phase("comm"):
            if self.wc.reducible:
                self.comm.reduce_sum(self.wire, dst=0)
            else:
                self.comm.gather(self.wire, self.gather_buf, dst=0)
        if self.is_master:
            with t.phase("decode"):
                okb = self.graph_split and self._graphB_step()
                if not okb:
                    if self.wc.reducible:
                        grad = self.wire
                    else:
                        self.agg.zero_()
                        self.wc.decode_all(self.gather_buf, self.agg)
                        grad = self.agg
                    self.opt.lr = self.lr
                    self._apply(grad)
        self._last_contrib = self.num_workers
        self._post_step()
        return self.last_loss

    def _step_body(self) -> None:
        """One full training step on static buffers (the captured body)."""
        self.flat_grad.zero_()
        loss = self.loss_fn(self.model(self._static_x), self._static_y)
        loss.backward()
        self._static_loss = loss.detach()
        if self.wc.reducible:
            grad = self.flat_grad
        else:
            self.wc.encode_all(self.wire, flat_grad=self.flat_grad)
            self.agg.zero_()
            self.wc.decode_all(self.wire.view(1, -1), self.agg)
            grad = self.agg
        self._apply(grad)

    def _whole_step_graphed(self, x: torch.Tensor, y: torch.Tensor) -> bool:
        """Capture-once/replay whole-step graph (lr and sampler seeds ride
        pinned->device copies, so replays track both).  Returns False (and
        disables itself) if capture fails — caller falls back to the
        standard path."""
        if self._wgraph is None:
            g = None
            try:
                self.opt.lr = self.lr
                self._lr_host[0] = self.lr
                self._static_x = x.clone()
                self._static_y = y.clone()
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    for _ in range(3):  # MIOpen algo search + warm states
                        self._step_body()
                torch.cuda.current_stream().wait_stream(side)
                g = torch.cuda.CUDAGraph()
                # thread_local: NCCL watchdog threads issue event queries
                # that would fail a global-mode capture at world > 1
                with torch.cuda.graph(g, capture_error_mode="thread_local"):
                    self._step_body()
                self._wgraph = g
            except Exception as exc:
                print(
                    f"[atomo] whole-step hipGraph capture failed ({exc}); "
                    "standard path",
                    flush=True,
                )
                self.graph_whole = False
                self._quarantine_graph(g)
                self._wgraph = None
                return False
        self._static_x.copy_(x)
        self._static_y.copy_(y)
        self._lr_host[0] = self.lr  # replay re-reads the pinned lr
        self.wc.advance_seeds()  # captured H2D copy re-reads pinned seeds
        self._wgraph.replay()
        if self.defer_loss:
            self._loss_tensor = self._static_loss
        else:
            self.last_loss = float(self._static_loss)
        return True

    def _fwd_bwd_graphed(self, x: torch.Tensor, y: torch.Tensor) -> None:
        if self._graph is None:
            try:
                self._static_x = x.clone()
                self._static_y = y.clone()
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    for _ in range(3):  # warmup: MIOpen algo selection
                        self.flat_grad.zero_()
                        loss = self.loss_fn(
                            self.model(self._static_x), self._static_y
                        )
                        loss.backward()
                torch.cuda.current_stream().wait_stream(side)
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g, capture_error_mode="thread_local"):
                    self.flat_grad.zero_()
                    self._static_loss = self.loss_fn(
                        self.model(self._static_x), self._static_y
                    )
                    self._static_loss.backward()
                self._graph = g
            except Exception as exc:  # capture-unsafe model: fall back
                print(f"[atomo] hipGraph capture failed ({exc}); eager fallback",
                      flush=True)
                self.use_graph = False
                self._graph = None
                self.flat_grad.zero_()
                loss = self.loss_fn(self.model(x), y)
                loss.backward()
                self.last_loss = float(loss.detach())
                return
        self._static_x.copy_(x)
        self._static_y.copy_(y)
        self._graph.replay()
        self.last_loss = float(self._static_loss)

    def _apply(self, grad_flat: torch.Tensor) -> None:
        scale = 1.0 / max(1, self._last_contrib or self.num_workers)
        if self.flat.is_cuda and type(self.opt).__name__ == "ExternalSGD":
            from ..ops import optim_ops

            lr_dev = None
            if torch.cuda.is_current_stream_capturing():
                # captured copy: replays re-read the pinned lr scalar
                self._lr_dev.copy_(self._lr_host, non_blocking=True)
                lr_dev = self._lr_dev
            optim_ops.fused_sgd(
                self.flat,
                grad_flat,
                self.opt.buf,
                lr=self.opt.lr,
                momentum=self.opt.momentum,
                weight_decay=self.opt.weight_decay,
                nesterov=self.opt.nesterov,
                dampening=self.opt.dampening,
                grad_scale=scale,
                lr_dev=lr_dev,
            )
        elif type(self.opt).__name__ == "ExternalAdam":
            self.opt.lr = self.lr
            self.opt.step(grad_flat, grad_scale=scale)
        else:
            if scale != 1.0:
                grad_flat = grad_flat * scale
            self.opt.step(grad_flat)

    # -----------------------------------------------------------------
    @torch.no_grad()
    def evaluate(self, loader, max_batches: int = 0) -> dict:
        """On-worker eval mirroring _evaluate_model
        (distributed_worker.py:344-370): loss + Prec@1/Prec@5."""
        from ..utils import accuracy

        self.model.eval()
        tot, loss_sum, p1_sum, p5_sum, nb = 0, 0.0, 0.0, 0.0, 0
        for i, (x, y) in enumerate(loader):
            if max_batches and i >= max_batches:
                break
            x, y = x.to(self.device), y.to(self.device)
            out = self.model(x)
            loss_sum += float(self.loss_fn(out, y)) * y.numel()
            k5 = min(5, out.shape[1])
            p1, p5 = accuracy(out, y, topk=(1, k5))
            p1_sum += p1 * y.numel()
            p5_sum += p5 * y.numel()
            tot += y.numel()
            nb += 1
        self.model.train()
        if tot == 0:
            return {"loss": float("nan"), "prec1": 0.0, "prec5": 0.0}
        return {"loss": loss_sum / tot, "prec1": p1_sum / tot, "prec5": p5_sum / tot}

    def save_checkpoint(self, path: Optional[str] = None) -> str:
        """torch.save to train_dir/model_step_<N> (reference
        sync_replicas_master_nn.py:331-336) plus optimizer/step state so a
        run can actually RESUME (the reference has no resume path)."""
        os.makedirs(self.train_dir, exist_ok=True)
        path = path or os.path.join(self.train_dir, f"model_step_{self.step_num}")
        torch.save(
            {
                "step": self.step_num,
                "lr": self.lr,
                "model": self.model.state_dict(),
                "optimizer": self.opt.state_dict() if self.opt else None,
            },
            path,
        )
        return path

    def load_checkpoint(self, path: str) -> None:
        """In-place resume: ``load_state_dict(assign=False)`` copies every
        tensor INTO the existing parameter views, so the flat buffer, the
        grad views, the WireCodec/encoder and any registered overlap hooks
        all stay valid — nothing is rebuilt.  (An earlier version rebuilt
        the WireCodec here, which orphaned the backward-hook overlap state
        and silently zeroed the SVD spectrum after resume.)"""
        ckpt = torch.load(path, map_location=self.device, weights_only=False)
        with torch.no_grad():
            self.model.load_state_dict(ckpt["model"], assign=False)
        self.flat_grad.zero_()
        self.step_num = ckpt["step"]
        self.lr = ckpt["lr"]
        if self.opt is not None and ckpt.get("optimizer") is not None:
            self.opt.load_state_dict(ckpt["optimizer"])
