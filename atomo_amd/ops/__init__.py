"""HIP/CDNA4 kernel bindings for MI355X (gfx950).

The extension ``_atomo_hip`` is built IN-TREE (``python setup.py build_ext
--inplace`` or ``__graft_entry__.build()``) so the .so travels with the repo
snapshot to GPU boxes.  Policy: on CUDA (= ROCm) tensors the HIP kernels are
the ONLY path — a missing extension raises instead of silently falling back
to eager PyTorch; CPU tensors use the torch oracle implementations in
atomo_amd.codings.

Kernels (sources in ops/csrc/, reference hot-spot map SURVEY §2.10):
  * qsgd_pack / qsgd_unpack_acc — one wave64 per bucket, shfl L2-norm
    reduction, hash-counter stochastic rounding, LDS bit-pack.
  * svd_decode_acc — fused u·diag(s)·vT rank-k reconstruction accumulated
    over ALL workers' packets in one output sweep.
  * fused_sgd — scale+weight-decay+momentum+apply in one flat sweep.
"""

from __future__ import annotations

import importlib
import os

_ext = None
_tried = False


def have_ext() -> bool:
    return _load(optional=True) is not None


def _load(optional: bool = False):
    global _ext, _tried
    if _ext is not None:
        return _ext
    if _tried and optional:
        return None
    _tried = True
    try:
        _ext = importlib.import_module("atomo_amd.ops._atomo_hip")
        return _ext
    except ImportError as e:
        if optional:
            return None
        raise RuntimeError(
            "atomo_amd HIP extension (_atomo_hip) is not built. Build it "
            "in-tree with `python setup.py build_ext --inplace` (or "
            "__graft_entry__.build()). Refusing to fall back to eager "
            "PyTorch on a GPU tensor."
        ) from e


def ext():
    return _load(optional=False)


from . import qsgd_ops, svd_ops, optim_ops  # noqa: E402,F401
