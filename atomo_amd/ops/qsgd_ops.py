"""QSGD pack/unpack GPU wrappers (kernels: ops/csrc/atomo_kernels.hip).

Wire layout per layer (inside an fp32 region, int words bit-cast):
    [norms (n_buckets fp32)] [packed (n_buckets * words_per_bucket) u32]
Element j of a bucket lives in word j // epw at bits (j % epw)*(1+q);
epw = 32 // (1+q).  Matches atomo_amd.codings.qsgd exactly (the CPU oracle).
"""

from __future__ import annotations

import torch

from . import ext

_seed_counter = [12345]


def _next_seed() -> int:
    _seed_counter[0] = (_seed_counter[0] * 6364136223846793005 + 1442695040888963407) % (
        1 << 63
    )
    return _seed_counter[0]


def set_seed(seed: int) -> None:
    _seed_counter[0] = int(seed) % (1 << 63)


def pack_into(
    flat: torch.Tensor,
    region: torch.Tensor,
    bucket_size: int,
    qlevel: int,
    scheme: str = "qsgd",
) -> None:
    assert flat.is_cuda and region.is_cuda
    ext().qsgd_pack(
        flat, region, int(bucket_size), int(qlevel), scheme == "terngrad", _next_seed()
    )


def unpack_accumulate(
    region: torch.Tensor,
    out: torch.Tensor,
    numel: int,
    bucket_size: int,
    qlevel: int,
) -> None:
    assert region.is_cuda and out.is_cuda
    ext().qsgd_unpack_acc(region, out, int(numel), int(bucket_size), int(qlevel))
