#!/usr/bin/env python3
"""Flagship benchmark: ResNet18 / CIFAR-100-shaped DDP training step.

Measures the reference's headline metric (seconds/epoch and
images/sec for ResNet18 CIFAR-100 DDP, global batch 256) on MI355X with
synthetic CIFAR-shaped data and random-init weights.

Contract: `python bench.py --gpus N --steps K --warmup W`; for N>1 the driver
launches it under torch.distributed.run with one rank per GPU (RCCL). W
untimed warmup steps, then exactly K timed steps bracketed by barrier +
torch.cuda.synchronize on both sides; elapsed is MAX over ranks; rank 0 prints
ONE JSON line.
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist

# Precision-matched reference rows (BASELINE.md, 4x2080Ti, 50k images/epoch):
# fp32 runs compare against the plain-DDP row (16 s/epoch = 3125 img/s);
# bf16/fp16 runs against the DDP+apex AMP row (14.5 s/epoch = 3448.3 img/s),
# reference README.md:76-77.
REFERENCE_IMAGES_PER_SEC = {"fp32": 3125.0, "bf16": 50000.0 / 14.5,
                            "fp16": 50000.0 / 14.5,
                            "bf16_o2": 50000.0 / 14.5}
REFERENCE_SECONDS_PER_EPOCH = {"fp32": 16.0, "bf16": 14.5, "fp16": 14.5,
                               "bf16_o2": 14.5}
EPOCH_IMAGES = 50000  # CIFAR-100 train set size


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--global-batch", "--global_batch", type=int, default=256)
    p.add_argument("--arch", type=str, default="resnet18")
    p.add_argument("--amp", type=str, default="bf16",
                   choices=["fp32", "bf16", "fp16", "bf16_o2"])
    p.add_argument("--mode", type=str, default="flat", choices=["flat", "torchddp"])
    p.add_argument("--grad-accu-steps", type=int, default=1)
    p.add_argument("--no-syncbn", action="store_true")
    p.add_argument("--channels-last", action="store_true", default=True,
                   help="NHWC + native implicit-GEMM conv kernels (default)")
    p.add_argument("--no-channels-last", dest="channels_last",
                   action="store_false")
    p.add_argument("--hip-graph", action="store_true")
    p.add_argument("--image-size", type=int, default=32)
    p.add_argument("--comm-bf16", action="store_true",
                   help="all-reduce gradients in bf16 (half the xGMI bytes)")
    return p.parse_args()


def main():
    args = parse_args()
    from mi355x_ddp.config import TrainConfig
    from mi355x_ddp.core import dist as dist_utils
    from mi355x_ddp.core.amp import autocast_ctx, build_scaler
    from mi355x_ddp.core.worker import build_training, init_seeds

    # --- distributed setup (torchrun env or single process) ---------------
    if "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1:
        rank = int(os.environ["RANK"])
        local_rank = int(os.environ["LOCAL_RANK"])
        world = int(os.environ["WORLD_SIZE"])
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend=backend)
    else:
        rank, local_rank, world = 0, 0, 1

    device = torch.device("cuda", local_rank) if torch.cuda.is_available() \
        else torch.device("cpu")
    init_seeds(1 + rank)

    cfg = TrainConfig(
        arch=args.arch, batch_size=args.global_batch, amp=args.amp,
        sync_bn=(not args.no_syncbn) and world > 1,
        grad_accu_steps=args.grad_accu_steps,
        channels_last=args.channels_last, hip_graph=args.hip_graph,
        comm_bf16=args.comm_bf16,
        use_flat_ddp=(args.mode == "flat"), synthetic=True)
    per_rank = cfg.per_rank_batch(world)

    model, criterion, optimizer, scheduler, scaler = build_training(
        cfg, device, world, rank, distributed=True,
        wrap="flat" if args.mode == "flat" else "torch")
    accu = max(1, args.grad_accu_steps)
    sub = max(1, per_rank // accu)

    # --- synthetic CIFAR-shaped batches, pinned host memory ---------------
    n_prefab = 8
    gen = torch.Generator().manual_seed(123 + rank)
    batches = []
    for _ in range(n_prefab):
        img = torch.randn(per_rank, 3, args.image_size, args.image_size,
                          generator=gen)
        lbl = torch.randint(0, 100, (per_rank,), generator=gen)
        if device.type == "cuda":
            img, lbl = img.pin_memory(), lbl.pin_memory()
        batches.append((img, lbl))

    graph_step = None
    if args.hip_graph:
        assert device.type == "cuda", "--hip-graph needs a GPU"
        assert args.mode == "flat" and accu == 1 and args.amp != "fp16"
        from mi355x_ddp.core.graphs import GraphedTrainStep
        model.train()
        graph_step = GraphedTrainStep(model, criterion, optimizer, cfg, device,
                                      batch=per_rank, image_size=args.image_size)

    def one_step(i: int):
        img_h, lbl_h = batches[i % n_prefab]
        if graph_step is not None:
            # static-buffer copy_ pulls straight from pinned host memory
            graph_step.run(img_h, lbl_h)
            return
        images = img_h.to(device, non_blocking=True)
        labels = lbl_h.to(device, non_blocking=True)
        if args.channels_last:
            images = images.to(memory_format=torch.channels_last)
        if hasattr(model, "zero_grad_buffer"):
            model.zero_grad_buffer()
        else:
            optimizer.zero_grad(set_to_none=False)
        for a in range(accu):
            s_img = images[a * sub:(a + 1) * sub] if accu > 1 else images
            s_lbl = labels[a * sub:(a + 1) * sub] if accu > 1 else labels
            is_last = a == accu - 1
            import contextlib
            ctx = model.no_sync() if (not is_last and hasattr(model, "no_sync")) \
                else contextlib.nullcontext()
            with ctx:
                with autocast_ctx(cfg.amp, device.type):
                    out = model(s_img)
                    loss = criterion(out, s_lbl) / accu
                (scaler.scale_loss(loss) if scaler else loss).backward()
        if hasattr(model, "finalize_backward"):
            model.finalize_backward()
        if scaler is not None:
            grads = [model.flat_grads] \
                if getattr(model, "flat_grads", None) is not None else \
                [p.grad for p in model.parameters() if p.grad is not None]
            scaler.unscale_(grads)
            scaler.step(optimizer)
        else:
            optimizer.step()
        flush_nbt()

    # --- warmup ------------------------------------------------------------
    from mi355x_ddp.ops.batchnorm import (
        defer_num_batches_tracked, flush_num_batches_tracked as flush_nbt)
    defer_num_batches_tracked(True)
    model.train()
    import sys
    for i in range(args.warmup):
        t_w = time.perf_counter()
        one_step(i)
        if device.type == "cuda":
            torch.cuda.synchronize()
        if rank == 0:
            print(f"[warmup {i}] {time.perf_counter() - t_w:.2f}s",
                  file=sys.stderr, flush=True)

    # --- timed region ------------------------------------------------------
    if world > 1:
        dist.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        one_step(i)
    if world > 1:
        dist.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if world > 1 and device.type == "cuda" else "cpu")
    if world > 1:
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    images_per_sec = args.steps * args.global_batch / elapsed
    if rank == 0:
        arch_label = {"resnet18": "ResNet18", "resnet34": "ResNet34",
                      "resnet50": "ResNet50", "resnet101": "ResNet101",
                      "resnet152": "ResNet152",
                      "resnet18_imagenet": "ResNet18-ImageNet",
                      "resnet50_imagenet": "ResNet50-ImageNet"}.get(
                          args.arch, args.arch)
        data_label = "CIFAR-100" if args.image_size == 32 else             f"synthetic-{args.image_size}px"
        result = {
            "metric": f"images/sec, {arch_label} {data_label} DDP training",
            "value": round(images_per_sec, 1),
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "strong",
            # the published baseline is ResNet18/CIFAR-100 only; other
            # configs have no reference number to compare against. The
            # divisor is precision-matched: AMP runs divide by the DDP+apex
            # row, fp32 runs by the plain-DDP row.
            "vs_baseline": round(
                images_per_sec / REFERENCE_IMAGES_PER_SEC[args.amp], 3)
            if (args.arch == "resnet18" and args.image_size == 32) else None,
            "dtype": args.amp,
            "data": "synthetic",
            "config": {
                "model": args.arch,
                "global_batch": args.global_batch,
                "seq_len": None,
                "image_size": args.image_size,
                "parallelism": f"dp{world}",
                "sync_bn": cfg.sync_bn,
                "grad_accu_steps": args.grad_accu_steps,
                "mode": args.mode,
                "seconds_per_epoch": round(EPOCH_IMAGES / images_per_sec, 4)
                if args.image_size == 32 else None,
                "reference_seconds_per_epoch_4x2080ti":
                REFERENCE_SECONDS_PER_EPOCH[args.amp]
                if (args.arch == "resnet18" and args.image_size == 32) else None,
            },
        }
        print(json.dumps(result), flush=True)
        if os.environ.get("MI355X_TUNE_REPORT", "0") == "1":
            from mi355x_ddp.ops.conv import tune_report
            print("[autotune choices]\n" + tune_report(), file=sys.stderr,
                  flush=True)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
