"""Wire-buffer encode/decode orchestration.

One fixed-layout fp32 "wire bucket" per rank carries every layer's coded
gradient (layout from Codec.build_specs).  The encoder fills a rank's bucket
from autograd grads; the decoder accumulates a stacked (W, words) gather of
buckets into the PS's flat fp32 aggregation buffer.

Replaces the reference's per-(layer, worker) pickled isend/irecv + numpy
aggregation (distributed_worker.py:313-335, sync_replicas_master_nn.py:281-329)
with device-resident tensors end to end.
"""

from __future__ import annotations

from typing import List

import torch

from ..codings import Codec, SVDCodec, QSGDCodec
from ..codings.base import LayerSpec


class WireCodec:
    """Binds a codec to a parameter list: layout, encode, decode."""

    def __init__(
        self,
        codec: Codec,
        params: List[torch.Tensor],
        device: torch.device,
        rank: int = 0,
    ):
        self.codec = codec
        self.params = params
        self.device = device
        # per-rank RNG decorrelation: every worker must make INDEPENDENT
        # stochastic-rounding / atom-selection draws (the reference's MPI
        # processes each have their own np.random state), otherwise the PS
        # average keeps ~one worker's compression variance instead of 1/W.
        self.rank = int(rank)
        rank_mix = self.rank * 0x9E3779B97F4A7C15
        self.specs = codec.build_specs([list(p.shape) for p in params])
        self.total_words = sum(s.wire_words for s in self.specs)
        self._batched_encoder = None
        # flat-parameter offsets (agg buffer layout == flat param layout)
        self.param_offsets = []
        off = 0
        for p in params:
            self.param_offsets.append(off)
            off += p.numel()
        self.total_params = off
        if (
            device.type == "cuda"
            and isinstance(codec, SVDCodec)
            and codec.compress
            and codec.backend in ("auto", "gram")
        ):
            from .svd_encoder import BatchedSVDEncoder

            self._batched_encoder = BatchedSVDEncoder(
                codec,
                self.specs,
                device,
                param_offsets=self.param_offsets,
                rank=self.rank,
            )
        # batched QSGD tables: one pack / unpack launch for the whole model
        self._qsgd_tables = None
        if device.type == "cuda" and isinstance(codec, QSGDCodec):
            from .. import ops
            from ..ops import qsgd_ops

            qsgd_ops.set_seed(12345 ^ rank_mix)
            if ops.have_ext():
                desc, pack_work, unpack_work = [], [], []
                for spec, p_off in zip(self.specs, self.param_offsets):
                    meta = spec.meta
                    nb, wpb = meta["n_buckets"], meta["words_per_bucket"]
                    li = len(desc)
                    desc.append([p_off, spec.numel, spec.wire_offset, nb, wpb])
                    for b in range(nb):
                        pack_work.append([li, b])
                    n_words = nb * wpb
                    for c in range((n_words + 255) // 256):
                        unpack_work.append([li, c])
                self._qsgd_tables = {
                    "desc": torch.tensor(desc, dtype=torch.int64, device=device),
                    "pack": torch.tensor(pack_work, dtype=torch.int32, device=device),
                    "unpack": torch.tensor(
                        unpack_work, dtype=torch.int32, device=device
                    ),
                }
                self._qsgd_seed = (987654321 ^ rank_mix) % (1 << 62)
                # device-resident sampler seed (refreshed via pinned host
                # memory) so hipGraph replays roll fresh rounding dice
                self._qsgd_seed_host = torch.zeros(
                    1, dtype=torch.int64, pin_memory=True
                )
                self._qsgd_seed_dev = torch.zeros(
                    1, dtype=torch.int64, device=device
                )
        # scratch for layers whose 2-D fold is zero-padded (odd 1-D sizes)
        self._pad_scratch = {}
        if isinstance(codec, SVDCodec) and codec.compress:
            for s in self.specs:
                if s.meta["padded"] != s.numel:
                    self._pad_scratch[s.index] = torch.zeros(
                        s.meta["padded"], dtype=torch.float32, device=device
                    )
        # batched SVD decode tables: one launch over all layers x workers
        self._svd_decode_tables = None
        if (
            device.type == "cuda"
            and isinstance(codec, SVDCodec)
            and codec.compress
        ):
            from .. import ops

            if ops.have_ext():
                DEC_CHUNK = 1024
                desc, work, kernel_layers = [], [], set()
                for s, p_off in zip(self.specs, self.param_offsets):
                    if s.meta["padded"] != s.numel:
                        continue  # odd folds decode through the scratch path
                    meta = s.meta
                    li = len(desc)
                    desc.append(
                        [p_off, meta["m"], meta["n"], meta["r_max"], s.wire_offset]
                    )
                    kernel_layers.add(s.index)
                    total = meta["m"] * meta["n"]
                    for c in range((total + DEC_CHUNK - 1) // DEC_CHUNK):
                        work.append([li, c])
                if desc:
                    self._svd_decode_tables = {
                        "desc": torch.tensor(desc, dtype=torch.int64, device=device),
                        "work": torch.tensor(work, dtype=torch.int32, device=device),
                        "layers": kernel_layers,
                    }

    @property
    def reducible(self) -> bool:
        return getattr(self.codec, "reducible", False)

    _device_counted = False

    def device_msg_bytes(self) -> float:
        """Msg bytes accumulated on device by the async sampler (one D2H)."""
        if self._batched_encoder is None or not self._device_counted:
            return 0.0
        return 4.0 * float(self._batched_encoder.used_words_dev.item())

    def reset_device_msg_bytes(self) -> None:
        if self._batched_encoder is not None and hasattr(
            self._batched_encoder, "used_words_dev"
        ):
            self._batched_encoder.used_words_dev.zero_()

    # -- overlap (backward-hook driven per-layer encode) -----------------
    # The reference prototypes communication/computation overlap with its
    # *Split models (per-layer backward interleaved with MPI Isend,
    # model_ops/resnet_split.py:539-575).  Here the same capability is a
    # post-accumulate-grad hook per parameter that launches that layer's
    # encode work on a side HIP stream while backward continues on the
    # deeper layers; encode_all(overlap_done=True) then skips the work.
    def setup_overlap(self, side_stream, params, wire) -> bool:
        import torch as _t

        if self.device.type != "cuda":
            return False
        self._overlap_handles = []
        if isinstance(self.codec, SVDCodec) and self._batched_encoder is not None:
            import os

            enc = self._batched_encoder
            enc.setup_solver_overlap(side_stream)
            # "big": hook only big-fold + host layers; j64 grams run as one
            # batched_gram launch post-backward.  "all": hook every layer
            # (the round-1 behavior).  Default "auto": big when big folds
            # exist (deep ResNets — hundreds of hooks cost more host time
            # than one kernel), else all (small models, where per-layer
            # grams fully hide under backward).
            mode = os.environ.get("ATOMO_OVERLAP_MODE", "auto")
            if mode == "auto":
                mode = "big" if enc.hook_layers else "all"
            enc.hook_mode = mode
            hook_set = (
                enc.hook_layers
                if enc.hook_mode == "big"
                else set(range(len(self.specs)))
            )

            def make_hook(i):
                spec = self.specs[i]
                sm = enc.small[i]
                gv = enc.grams[
                    enc.gram_offsets[i] : enc.gram_offsets[i] + sm * sm
                ].view(sm, sm)
                tall_is_m = enc.m_is_tall[i]

                def hook(p):
                    evt = _t.cuda.Event()
                    evt.record()
                    with _t.cuda.stream(side_stream):
                        side_stream.wait_event(evt)
                        a = enc._a2d(p.grad, spec)
                        if tall_is_m:
                            _t.mm(a.t(), a, out=gv)
                        else:
                            _t.mm(a, a.t(), out=gv)
                        enc.on_overlap_gram_done(i)

                return hook

            for i, p in enumerate(params):
                if i in hook_set:
                    self._overlap_handles.append(
                        p.register_post_accumulate_grad_hook(make_hook(i))
                    )
            return True
        if isinstance(self.codec, QSGDCodec):
            from ..ops import qsgd_ops

            codec = self.codec

            def make_hook(i):
                spec = self.specs[i]
                region = wire[spec.wire_offset : spec.wire_offset + spec.wire_words]

                def hook(p):
                    evt = _t.cuda.Event()
                    evt.record()
                    with _t.cuda.stream(side_stream):
                        side_stream.wait_event(evt)
                        qsgd_ops.pack_into(
                            p.grad.reshape(-1),
                            region,
                            codec.bucket_size,
                            codec.qlevel,
                            codec.scheme,
                        )

                return hook

            for i, p in enumerate(params):
                self._overlap_handles.append(
                    p.register_post_accumulate_grad_hook(make_hook(i))
                )
            return True
        return False

    def remove_overlap(self) -> None:
        """Unregister every backward hook installed by setup_overlap."""
        for h in getattr(self, "_overlap_handles", []):
            h.remove()
        self._overlap_handles = []

    def advance_seeds(self) -> None:
        """Host-side LCG step for whichever device sampler this codec
        uses; under a whole-step/split hipGraph the trainer calls this
        before each replay (the captured pinned-memory copy re-reads the
        scalar at replay time)."""
        if self._batched_encoder is not None:
            self._batched_encoder.advance_seed()
        if self._qsgd_tables is not None:
            self._qsgd_seed = (
                self._qsgd_seed * 6364136223846793005 + 1442695040888963407
            ) % (1 << 62)
            self._qsgd_seed_host[0] = self._qsgd_seed

    def arm_overlap(self) -> None:
        if self._batched_encoder is not None:
            self._batched_encoder.arm_overlap()

    # -- worker side -----------------------------------------------------
    def encode_all(
        self,
        wire: torch.Tensor,
        flat_grad: torch.Tensor = None,
        overlap_done: bool = False,
    ) -> int:
        """Encode every parameter's .grad into ``wire``; returns fp32 words
        actually used (the Msg bytes counter).  overlap_done=True means the
        backward hooks already did the per-layer front half (QSGD: the whole
        pack; SVD: the Gram matrices)."""
        if overlap_done and isinstance(self.codec, QSGDCodec):
            return sum(s.wire_words for s in self.specs)
        if self._qsgd_tables is not None and flat_grad is not None:
            import torch as _t

            from ..ops import ext

            t = self._qsgd_tables
            if _t.cuda.is_current_stream_capturing():
                # captured H2D copy re-reads the pinned scalar per replay
                self._qsgd_seed_dev.copy_(
                    self._qsgd_seed_host, non_blocking=True
                )
            else:
                self.advance_seeds()
                self._qsgd_seed_dev.fill_(self._qsgd_seed)
            ext().qsgd_pack_batched(
                flat_grad, wire, t["desc"], t["pack"], t["pack"].shape[0],
                self.codec.bucket_size, self.codec.qlevel,
                self.codec.scheme == "terngrad", self._qsgd_seed_dev,
            )
            return sum(s.wire_words for s in self.specs)
        grads = [
            p.grad if p.grad is not None else torch.zeros_like(p) for p in self.params
        ]
        if self._batched_encoder is not None:
            used = self._batched_encoder.encode_all(
                grads, wire, flat_grad=flat_grad, grams_done=overlap_done
            )
            if getattr(self._batched_encoder, "device_counted", False):
                self._device_counted = True
            return used
        used = 0
        for grad, spec in zip(grads, self.specs):
            region = wire[spec.wire_offset : spec.wire_offset + spec.wire_words]
            used += self.codec.encode_into(grad, region, spec)
        return used

    # -- master side -----------------------------------------------------
    def decode_all(self, stacked: torch.Tensor, agg: torch.Tensor) -> None:
        """agg (flat, total_params) += sum over rows of ``stacked``."""
        W = stacked.shape[0]
        use_hip = stacked.is_cuda
        if self._qsgd_tables is not None:
            from ..ops import ext

            t = self._qsgd_tables
            for w in range(W):
                ext().qsgd_unpack_batched(
                    stacked[w].contiguous()
                    if not stacked[w].is_contiguous()
                    else stacked[w],
                    agg, t["desc"], t["unpack"], t["unpack"].shape[0],
                    self.codec.bucket_size, self.codec.qlevel,
                )
            return
        dec_tables = self._svd_decode_tables if use_hip else None
        if dec_tables is not None:
            from ..ops import ext

            ext().svd_decode_batched(
                stacked, agg, dec_tables["desc"], dec_tables["work"],
                dec_tables["work"].shape[0],
            )
        covered = dec_tables["layers"] if dec_tables is not None else set()
        for spec, p_off in zip(self.specs, self.param_offsets):
            if spec.index in covered:
                continue
            regions = stacked.narrow(1, spec.wire_offset, spec.wire_words)
            out = agg[p_off : p_off + spec.numel]
            if isinstance(self.codec, SVDCodec) and self.codec.compress:
                meta = spec.meta
                if use_hip:
                    from ..ops import svd_ops

                    if spec.index in self._pad_scratch:
                        scratch = self._pad_scratch[spec.index]
                        scratch.zero_()
                        svd_ops.decode_acc(
                            regions,
                            scratch.view(meta["m"], meta["n"]),
                            meta["m"],
                            meta["n"],
                            meta["r_max"],
                        )
                        out += scratch[: spec.numel]
                    else:
                        svd_ops.decode_acc(
                            regions,
                            out.view(meta["m"], meta["n"]),
                            meta["m"],
                            meta["n"],
                            meta["r_max"],
                        )
                else:
                    for w in range(W):
                        self.codec.decode_from(regions[w], out, spec)
            else:
                for w in range(W):
                    self.codec.decode_from(regions[w], out, spec)
