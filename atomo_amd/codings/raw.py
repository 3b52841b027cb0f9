"""Raw fp32 pass-through codec (the ``--code=sgd`` path).

The reference routes ``--code=sgd`` through a missing ``lossless_compress``
module wrapping blosc on the host (sync_replicas_master_nn.py:134-138,
src/utils.py:1-16).  On MI355X a host compression hop would cost more than it
saves over 153 GB/s xGMI links, so this codec is a straight device-resident
copy; the PS runtime additionally short-circuits it to a single RCCL
``reduce`` when every rank uses it (no gather needed at all).
"""

from __future__ import annotations

from typing import Dict, Sequence

import torch

from .base import Codec, LayerSpec


class RawCodec(Codec):
    name = "sgd"

    def __init__(self, **kwargs):
        pass

    # reduce-friendly: the wire region IS the gradient
    reducible = True

    def wire_words(self, shape: Sequence[int]) -> int:
        return int(torch.Size(shape).numel())

    def encode(self, grad: torch.Tensor) -> Dict:
        return {"grad": grad, "encode": False}

    def decode(self, code: Dict) -> torch.Tensor:
        return code["grad"]

    def encode_into(self, grad: torch.Tensor, region: torch.Tensor, spec: LayerSpec) -> int:
        region[: spec.numel].copy_(grad.reshape(-1))
        return spec.numel

    def decode_from(self, region: torch.Tensor, out: torch.Tensor, spec: LayerSpec) -> None:
        out += region[: spec.numel]
