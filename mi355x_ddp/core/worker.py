"""main_worker — the one vertical slice every entry script drives.

Keeps the reference's sacred signature main_worker(local_rank, nprocs, cfg)
(reference README.md:100-110, distributed.py:33-41) over the new runtime:
RCCL process group, fused-kernel model, FlatDDP (or torch DDP) wrap, FusedSGD,
MultiStepLR, sharded synthetic/real CIFAR loaders, engine.fit.
"""
from __future__ import annotations

import os
import random
from typing import Optional

import numpy as np
import torch

from ..config import TrainConfig
from ..data import build_loaders
from ..models import build_model
from ..ops import FusedSGD
from ..ops.batchnorm import MI355SyncBatchNorm
from ..ops.xent import SoftmaxCrossEntropy
from ..parallel import FlatDDP, wrap_torch_ddp
from . import dist as dist_utils
from .amp import build_scaler
from .checkpoint import load_checkpoint
from .engine import fit


def init_seeds(seed: int, deterministic: bool = False,
               benchmark: bool = False) -> None:
    """Per-rank seeding (reference distributed_mp.py:29-39), without the
    reference's quirk of re-enabling benchmark mode right after requesting
    determinism (distributed_mp.py:61).

    `benchmark` defaults OFF (the reference set cudnn.benchmark=True,
    distributed.py:48): on ROCm that flag switches MIOpen to exhaustive
    find, which COMPILES candidate kernels per conv shape — minutes per
    224px shape on a cold cache, measured hanging the first training step.
    The per-shape autotune cache in ops/conv.py is this framework's
    benchmark=True equivalent (it also covers the MIOpen immediate-mode
    kernel as a candidate)."""
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    torch.backends.cudnn.deterministic = deterministic
    torch.backends.cudnn.benchmark = benchmark and not deterministic


def build_training(cfg: TrainConfig, device: torch.device, world_size: int,
                   rank: int, distributed: bool = True, wrap: str = "flat"):
    """Model + criterion + optimizer + scheduler + scaler, wrapped for DP mode."""
    model = build_model(cfg.arch, cfg.num_classes).to(device)
    if cfg.native_ops:
        from ..ops import MI355Conv2d
        from ..ops.pool import MI355MaxPool2d
        model = MI355Conv2d.convert(model)
        model = MI355MaxPool2d.convert(model)
    if cfg.channels_last:
        model = model.to(memory_format=torch.channels_last)
    if distributed and world_size > 1 and cfg.sync_bn:
        model = MI355SyncBatchNorm.convert_sync_batchnorm(model)
    if cfg.amp == "bf16_o2":
        from .amp import cast_model_bf16
        model = cast_model_bf16(model)
    if distributed and world_size >= 1 and wrap == "flat":
        model = FlatDDP(model, bucket_cap_mb=cfg.bucket_cap_mb,
                        overlap=not cfg.hip_graph,
                        # O2: the whole model (and its grads) is bf16, so the
                        # flat buffer and the wire are bf16 natively
                        grad_dtype=torch.bfloat16 if cfg.amp == "bf16_o2"
                        else torch.float32,
                        comm_dtype=torch.bfloat16 if cfg.comm_bf16 else None,
                        # hipGraph capture needs static grad memory even at
                        # world 1; otherwise world 1 skips the flat buffer
                        static_grads=True if cfg.hip_graph else None)
    elif distributed and wrap == "torch":
        model = wrap_torch_ddp(model,
                               device.index if device.type == "cuda" else None,
                               bucket_cap_mb=cfg.bucket_cap_mb)
    criterion = SoftmaxCrossEntropy().to(device)
    optimizer = FusedSGD(model.parameters(), lr=cfg.lr, momentum=cfg.momentum,
                         weight_decay=cfg.weight_decay)
    scheduler = build_scheduler(optimizer, cfg)
    scaler = build_scaler(cfg.amp)
    return model, criterion, optimizer, scheduler, scaler


def build_scheduler(optimizer, cfg: TrainConfig):
    """MultiStepLR (the reference's schedule, distributed.py:64) or cosine,
    optionally behind a linear warmup."""
    sched = getattr(cfg, "lr_schedule", "multistep")
    warm = max(0, getattr(cfg, "warmup_epochs", 0))
    if sched == "cosine":
        main = torch.optim.lr_scheduler.CosineAnnealingLR(
            optimizer, T_max=max(1, cfg.epochs - warm))
    else:
        main = torch.optim.lr_scheduler.MultiStepLR(
            optimizer, milestones=cfg.lr_milestones, gamma=cfg.lr_gamma)
    if warm == 0:
        return main
    warmup = torch.optim.lr_scheduler.LinearLR(
        optimizer, start_factor=1.0 / (warm + 1), total_iters=warm)
    return torch.optim.lr_scheduler.SequentialLR(
        optimizer, [warmup, main], milestones=[warm])


def main_worker(local_rank: int, nprocs: int, cfg: TrainConfig,
                rank: Optional[int] = None, init_pg: bool = True) -> float:
    rank = local_rank if rank is None else rank
    if init_pg and nprocs >= 1:
        dist_utils.init_distributed(rank, nprocs, cfg.ip, cfg.port,
                                    backend=cfg.backend, device_id=local_rank,
                                    timeout_s=cfg.dist_timeout_s)
    init_seeds(cfg.seed + rank + 1)
    device = torch.device("cuda", local_rank) if torch.cuda.is_available() \
        else torch.device("cpu")
    wrap = "flat" if cfg.use_flat_ddp else "torch"
    model, criterion, optimizer, scheduler, scaler = build_training(
        cfg, device, nprocs, rank, distributed=True, wrap=wrap)
    start_epoch = 0
    if cfg.resume:
        state = load_checkpoint(cfg.resume, model, optimizer, scheduler, scaler,
                                map_location=device)
        start_epoch = state["epoch"] + 1
        if dist_utils.is_main_process():
            print(f"resumed from {cfg.resume} at epoch {start_epoch}")
    train_loader, test_loader, train_sampler = build_loaders(
        cfg, nprocs, rank, distributed=nprocs > 1)
    if cfg.evaluate:
        from .engine import validate
        best = validate(model, test_loader, criterion, device, cfg)
        dist_utils.cleanup()
        return best
    best = fit(model, train_loader, test_loader, train_sampler, criterion,
               optimizer, scheduler, cfg, device, scaler=scaler,
               start_epoch=start_epoch)
    dist_utils.cleanup()
    return best
