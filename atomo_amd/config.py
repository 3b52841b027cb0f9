"""CLI flag surface and typed run configuration.

Preserves the reference's flag set (distributed_nn.py:31-82) so existing
launch scripts port directly, and fixes its ``type=bool`` trap
(``--enable-gpu=`` meaning False, distributed_nn.py:70-75) by using proper
boolean flags that ALSO accept the reference's ``--enable-gpu=<anything>``
spelling.  New flags are additive: --optimizer, --dedicated-ps,
--shrink-freq, --checkpoint-freq, --weight-decay, --svd-backend, --overlap.
"""

from __future__ import annotations

import argparse
import dataclasses
from typing import Optional

import torch

from .codings import make_codec
from .data import dataset_spec


def _ref_bool(v: str) -> bool:
    # reference semantics: any non-empty string is truthy
    if isinstance(v, bool):
        return v
    return bool(v) and v.lower() not in ("0", "false", "no", "off", "")


def add_fit_args(parser: argparse.ArgumentParser) -> argparse.ArgumentParser:
    p = parser
    p.add_argument("--batch-size", type=int, default=128)
    p.add_argument("--test-batch-size", type=int, default=500)
    p.add_argument("--max-steps", type=int, default=10000)
    p.add_argument("--epochs", type=int, default=100)
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--momentum", type=float, default=0.9)
    p.add_argument("--lr-shrinkage", type=float, default=0.95)
    p.add_argument("--no-cuda", action="store_true", default=False)
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--log-interval", type=int, default=10)
    p.add_argument("--network", type=str, default="ResNet18")
    p.add_argument("--code", type=str, default="svd", choices=["sgd", "svd", "qsgd"])
    p.add_argument("--bucket-size", type=int, default=512)
    p.add_argument("--dataset", type=str, default="cifar10")
    p.add_argument("--data-root", type=str, default=None,
                   help="directory with the real dataset files (standard "
                        "binary formats); synthetic data when absent")
    p.add_argument("--comm-type", type=str, default="Bcast",
                   help="Bcast = collective gather; P2P = arrival-order "
                        "isend/irecv with decode-as-arrives")
    p.add_argument("--num-aggregate", type=int, default=0,
                   help="with --comm-type P2P: aggregate only the first K "
                        "arriving gradients (0 = all; the reference stores "
                        "this flag but never implements it)")
    p.add_argument("--eval-freq", type=int, default=0)
    p.add_argument("--train-dir", type=str, default="output/models/")
    p.add_argument("--compress", type=_ref_bool, nargs="?", const=True, default=True)
    p.add_argument("--enable-gpu", type=_ref_bool, nargs="?", const=True, default=None)
    p.add_argument("--svd-rank", type=int, default=3)
    p.add_argument("--quantization-level", type=int, default=4)
    # new flags
    p.add_argument("--optimizer", type=str, default="sgd", choices=["sgd", "adam"])
    p.add_argument("--weight-decay", type=float, default=0.0)
    p.add_argument("--dedicated-ps", action="store_true", default=False,
                   help="rank 0 is a decode/apply-only PS (reference topology); "
                        "default colocates a worker on rank 0")
    p.add_argument("--shrink-freq", type=int, default=50)
    p.add_argument("--checkpoint-freq", type=int, default=0)
    p.add_argument("--wire-dtype", type=str, default="fp32",
                   choices=["fp32", "bf16"],
                   help="comm dtype: bf16 halves weight-push and svd-packet "
                        "bytes over xGMI (fp32 default; invalid for qsgd)")
    p.add_argument("--resume", type=str, default=None, nargs="?", const="latest",
                   help="checkpoint path to resume from, or 'latest' to pick "
                        "the newest model_step_<N> in --train-dir")
    p.add_argument("--svd-backend", type=str, default="auto",
                   choices=["auto", "torch", "gram"])
    p.add_argument("--overlap", action="store_true", default=False,
                   help="encode layers on a side stream as backward produces them")
    p.add_argument("--graph", action="store_true", default=False,
                   help="capture forward/backward in a hipGraph and replay")
    p.add_argument("--amp", action="store_true", default=False,
                   help="bf16 autocast for forward/backward (fp32 grads, "
                        "codec and optimizer unchanged)")
    p.add_argument("--step-timeout", type=float, default=0.0,
                   help="abort the rank if a global step stalls this many "
                        "seconds (straggler/hang watchdog; 0 = off)")
    return parser


@dataclasses.dataclass
class RunConfig:
    args: argparse.Namespace

    @property
    def device(self) -> torch.device:
        want_gpu = self.args.enable_gpu
        if want_gpu is None:
            want_gpu = not self.args.no_cuda
        if want_gpu and torch.cuda.is_available():
            return torch.device("cuda")
        return torch.device("cpu")

    def build_codec(self, generator: Optional[torch.Generator] = None):
        a = self.args
        return make_codec(
            a.code,
            rank=a.svd_rank,
            random_sample=True,
            compress=a.compress,
            backend=a.svd_backend,
            quantization_level=a.quantization_level,
            bucket_size=a.bucket_size,
            generator=generator,
        )

    def trainer_kwargs(self):
        a = self.args
        spec = dataset_spec(a.dataset)
        return dict(
            model_name=a.network,
            lr=a.lr,
            momentum=a.momentum,
            weight_decay=a.weight_decay,
            optimizer=a.optimizer,
            lr_shrinkage=a.lr_shrinkage,
            shrink_freq=a.shrink_freq,
            num_classes=spec["classes"],
            in_channels=spec["shape"][0],
            dedicated_ps=a.dedicated_ps,
            seed=a.seed,
            checkpoint_freq=a.checkpoint_freq,
            train_dir=a.train_dir,
            use_graph=a.graph,
            overlap=a.overlap,
            comm_type=a.comm_type,
            num_aggregate=a.num_aggregate,
            step_timeout=a.step_timeout,
            amp=a.amp,
            wire_dtype=a.wire_dtype,
        )


def parse_args(argv=None) -> RunConfig:
    parser = argparse.ArgumentParser(description="atomo_amd PS training")
    add_fit_args(parser)
    return RunConfig(parser.parse_args(argv))
