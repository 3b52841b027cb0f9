"""Codec base class and the fixed-layout wire contract.

The reference ships each layer's coded gradient as a pickled dict over MPI
(distributed_worker.py:313-335).  Here every layer gets a *fixed* fp32 region
inside one flat device tensor (the "wire buffer"), so a whole step's gradients
move in a single RCCL gather with no serialization: variable-rank SVD packets
write a run-length header word into their region, QSGD regions are exactly
sized, raw regions are the gradient itself.
"""

from __future__ import annotations

import dataclasses
from typing import Dict, List, Sequence

import torch


@dataclasses.dataclass(frozen=True)
class LayerSpec:
    """Static per-layer wire layout, computed once at model build."""

    index: int
    shape: tuple  # original parameter shape
    numel: int
    wire_offset: int  # offset (in fp32 words) into the shared wire buffer
    wire_words: int  # fixed region size in fp32 words
    meta: dict = dataclasses.field(default_factory=dict)


class Codec:
    """Interface every gradient codec implements.

    ``encode``/``decode`` are the dict-based oracle API (tests, single-layer
    use).  ``encode_into``/``decode_from`` are the hot path: encode straight
    into a pre-allocated wire region / decode-and-accumulate straight into the
    flat fp32 aggregation buffer, never leaving the device.
    """

    name = "base"

    # -- layout ----------------------------------------------------------
    def wire_words(self, shape: Sequence[int]) -> int:
        """Fixed fp32-word budget for one layer of this shape."""
        raise NotImplementedError

    def build_specs(self, shapes: List[Sequence[int]]) -> List[LayerSpec]:
        specs, off = [], 0
        for i, shp in enumerate(shapes):
            w = self.wire_words(shp)
            specs.append(
                LayerSpec(
                    index=i,
                    shape=tuple(shp),
                    numel=int(torch.Size(shp).numel()),
                    wire_offset=off,
                    wire_words=w,
                    meta=self.layer_meta(shp),
                )
            )
            off += w
        return specs

    def layer_meta(self, shape: Sequence[int]) -> dict:
        return {}

    # -- oracle API ------------------------------------------------------
    def encode(self, grad: torch.Tensor) -> Dict:
        raise NotImplementedError

    def decode(self, code: Dict) -> torch.Tensor:
        raise NotImplementedError

    # -- wire API --------------------------------------------------------
    def encode_into(self, grad: torch.Tensor, region: torch.Tensor, spec: LayerSpec) -> int:
        """Encode ``grad`` into ``region`` (a view of the wire buffer).

        Returns the number of fp32 words actually used (<= spec.wire_words);
        the runtime sums this into the grad-bytes/step counter, mirroring the
        reference's ``Msg(MB)`` metric (distributed_worker.py:326).
        """
        raise NotImplementedError

    def decode_from(self, region: torch.Tensor, out: torch.Tensor, spec: LayerSpec) -> None:
        """Decode ``region`` and ACCUMULATE (+=) into ``out`` (flat fp32,
        numel == spec.numel), mirroring aggregate_gradient
        (sync_replicas_master_nn.py:292-296) without the host round trip."""
        raise NotImplementedError

    # round-trip through the wire path, for tests
    def roundtrip(self, grad: torch.Tensor) -> torch.Tensor:
        spec = self.build_specs([list(grad.shape)])[0]
        wire = torch.zeros(spec.wire_words, dtype=torch.float32, device=grad.device)
        self.encode_into(grad, wire, spec)
        out = torch.zeros(spec.numel, dtype=torch.float32, device=grad.device)
        self.decode_from(wire, out, spec)
        return out.view(spec.shape)
