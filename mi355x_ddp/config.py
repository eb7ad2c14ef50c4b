"""Single config object consumed by every entry point.

The reference scattered ~8 argparse flags across six scripts and shipped a dead
``utils/config.py`` (reference utils/config.py:1-10 — imported by nothing). Here
the dataclass is the one source of truth; the reference's intended-but-unwired
knobs (save_epoch, tensorboard/metrics dir) are real.
"""
from __future__ import annotations

import argparse
import dataclasses
import os
from dataclasses import dataclass, field
from typing import List, Optional

# CIFAR-100 channel statistics (reference utils/dataset.py:8 hard-codes these).
CIFAR100_MEAN = (0.5070751592371323, 0.48654887331495095, 0.4409178433670343)
CIFAR100_STD = (0.2673342858792401, 0.2564384629170883, 0.27615047132568404)


@dataclass
class TrainConfig:
    # model / task
    arch: str = "resnet18"            # resnet18 | resnet34 | resnet50
    num_classes: int = 100
    image_size: int = 32

    # optimisation (reference defaults: distributed.py:18-25,63-64)
    batch_size: int = 256             # GLOBAL batch, divided by world size
    epochs: int = 200
    lr: float = 0.1
    momentum: float = 0.9
    weight_decay: float = 1e-4
    lr_milestones: List[int] = field(default_factory=lambda: [60, 120, 160])
    lr_gamma: float = 0.2
    # scheduler: "multistep" (reference default, distributed.py:64) or
    # "cosine"; warmup_epochs > 0 prepends a linear LR warmup (beyond the
    # reference, standard large-batch practice)
    lr_schedule: str = "multistep"
    warmup_epochs: int = 0
    grad_accu_steps: int = 1          # micro-batching; collectives elided on non-final micro-steps

    # precision: "fp32" | "bf16" | "fp16" | "bf16_o2"
    # (fp16 uses the dynamic loss scaler; bf16_o2 = apex-O2-equivalent pure
    # bf16 model with fp32 master weights in the optimizer)
    amp: str = "fp32"

    # distributed
    ip: str = "127.0.0.1"
    port: int = 23456
    backend: Optional[str] = None     # None -> nccl(RCCL) when a GPU is visible, else gloo
    sync_bn: bool = True              # reference enables SyncBN in every DDP entry
    # DDP gradient bucket size; None -> parallel.flat_ddp.bucket_cap_for()
    # picks by world size / model size (xGMI latency-vs-overlap policy)
    bucket_cap_mb: Optional[float] = None
    comm_bf16: bool = False           # all-reduce gradients in bf16 (half the wire bytes)
    dist_timeout_s: int = 600         # collective timeout (a dead rank fails fast)
    # stall diagnostics: dump all-thread stacks to stderr if one training
    # step exceeds this many seconds (None = off); see core/watchdog.py
    stall_dump_s: Optional[float] = 300.0
    use_flat_ddp: bool = True         # MI355X-native flat-bucket reducer (graph-capturable)

    # data
    data_root: str = "./data"
    synthetic: bool = True            # no network in this environment; real CIFAR if present
    num_workers: int = 4
    pin_memory: bool = True
    seed: int = 0

    # native kernels
    native_ops: bool = True           # HIP kernels when extension present; fail loudly on GPU if absent
    channels_last: bool = False
    hip_graph: bool = False           # capture the train step in a hipGraph

    # logging / checkpointing (makes reference utils/config.py's dead knobs real)
    log_interval: int = 10            # steps between loss all-reduce + print (reference did it every step)
    metrics_dir: str = "runs"
    ckpt_dir: str = "ckpts"
    save_epoch: int = 15
    eval_every_epoch: bool = True
    resume: str = ""
    evaluate: bool = False            # validation only (with --resume)
    # step caps (None = full epoch); used by smoke tests and quick CLI runs
    max_train_steps: Optional[int] = None
    max_eval_steps: Optional[int] = None

    def per_rank_batch(self, world_size: int) -> int:
        return max(1, self.batch_size // max(1, world_size))

    def replace(self, **kw) -> "TrainConfig":
        return dataclasses.replace(self, **kw)


def add_common_args(p: argparse.ArgumentParser) -> argparse.ArgumentParser:
    """Reference-compatible flag surface (reference distributed.py:18-25)."""
    p.add_argument("--seed", default=0, type=int, help="seed for initializing training")
    p.add_argument("--batch_size", "--batch-size", default=256, type=int,
                   help="global batch size across all GPUs")
    p.add_argument("--epochs", default=200, type=int)
    p.add_argument("--lr", "--learning-rate", default=0.1, type=float)
    p.add_argument("--ip", default="127.0.0.1", type=str)
    p.add_argument("--port", default=23456, type=int)
    p.add_argument("--arch", default="resnet18", type=str,
                   choices=["resnet18", "resnet34", "resnet50", "resnet101",
                            "resnet152", "resnet18_imagenet",
                            "resnet50_imagenet"])
    p.add_argument("--amp", default=None, type=str,
                   choices=[None, "fp32", "bf16", "fp16", "bf16_o2"])
    p.add_argument("--no-sync-bn", action="store_true")
    p.add_argument("--num_workers", default=4, type=int)
    p.add_argument("--log_interval", default=10, type=int)
    p.add_argument("--data_root", default="./data", type=str)
    p.add_argument("--ckpt_dir", default="ckpts", type=str)
    p.add_argument("--metrics_dir", default="runs", type=str)
    p.add_argument("--save_epoch", default=15, type=int)
    p.add_argument("--resume", default="", type=str)
    p.add_argument("--synthetic", action="store_true", default=None,
                   help="force synthetic CIFAR-shaped data (default: auto)")
    p.add_argument("--max_train_steps", default=None, type=int,
                   help="cap steps per epoch (smoke runs)")
    p.add_argument("--max_eval_steps", default=None, type=int)
    p.add_argument("--dist_timeout_s", default=None, type=int,
                   help="process-group collective timeout in seconds")
    p.add_argument("--lr_schedule", default=None, type=str,
                   choices=[None, "multistep", "cosine"])
    p.add_argument("--channels-last", "--channels_last", dest="channels_last",
                   action="store_true", default=None,
                   help="NHWC layout + native implicit-GEMM conv kernels")
    p.add_argument("--no-channels-last", dest="channels_last",
                   action="store_false")
    p.add_argument("--warmup_epochs", default=None, type=int)
    p.add_argument("--evaluate", action="store_true",
                   help="run validation only (use with --resume)")
    return p


def config_from_args(args: argparse.Namespace, **overrides) -> TrainConfig:
    cfg = TrainConfig()
    mapping = dict(
        seed="seed", batch_size="batch_size", epochs="epochs", lr="lr", ip="ip",
        port="port", arch="arch", num_workers="num_workers", log_interval="log_interval",
        data_root="data_root", ckpt_dir="ckpt_dir", metrics_dir="metrics_dir",
        save_epoch="save_epoch", resume="resume",
    )
    kw = {}
    for argname, cfgname in mapping.items():
        if hasattr(args, argname) and getattr(args, argname) is not None:
            kw[cfgname] = getattr(args, argname)
    if getattr(args, "amp", None):
        kw["amp"] = args.amp
    if getattr(args, "no_sync_bn", False):
        kw["sync_bn"] = False
    if getattr(args, "synthetic", None) is not None:
        kw["synthetic"] = bool(args.synthetic)
    if getattr(args, "grad_accu_steps", None):
        kw["grad_accu_steps"] = args.grad_accu_steps
    if getattr(args, "max_train_steps", None) is not None:
        kw["max_train_steps"] = args.max_train_steps
    if getattr(args, "max_eval_steps", None) is not None:
        kw["max_eval_steps"] = args.max_eval_steps
    if getattr(args, "dist_timeout_s", None) is not None:
        kw["dist_timeout_s"] = args.dist_timeout_s
    if getattr(args, "lr_schedule", None):
        kw["lr_schedule"] = args.lr_schedule
    if getattr(args, "warmup_epochs", None) is not None:
        kw["warmup_epochs"] = args.warmup_epochs
    if getattr(args, "evaluate", False):
        kw["evaluate"] = True
    if getattr(args, "channels_last", None) is not None:
        kw["channels_last"] = bool(args.channels_last)
    kw.update(overrides)
    return TrainConfig(**kw)
