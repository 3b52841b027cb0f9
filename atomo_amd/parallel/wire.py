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

    def __init__(self, codec: Codec, params: List[torch.Tensor], device: torch.device):
        self.codec = codec
        self.params = params
        self.device = device
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
                codec, self.specs, device, param_offsets=self.param_offsets
            )
        # scratch for layers whose 2-D fold is zero-padded (odd 1-D sizes)
        self._pad_scratch = {}
        if isinstance(codec, SVDCodec) and codec.compress:
            for s in self.specs:
                if s.meta["padded"] != s.numel:
                    self._pad_scratch[s.index] = torch.zeros(
                        s.meta["padded"], dtype=torch.float32, device=device
                    )

    @property
    def reducible(self) -> bool:
        return getattr(self.codec, "reducible", False)

    # -- worker side -----------------------------------------------------
    def encode_all(self, wire: torch.Tensor, flat_grad: torch.Tensor = None) -> int:
        """Encode every parameter's .grad into ``wire``; returns fp32 words
        actually used (the Msg bytes counter)."""
        grads = [
            p.grad if p.grad is not None else torch.zeros_like(p) for p in self.params
        ]
        if self._batched_encoder is not None:
            return self._batched_encoder.encode_all(grads, wire, flat_grad=flat_grad)
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
        for spec, p_off in zip(self.specs, self.param_offsets):
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
