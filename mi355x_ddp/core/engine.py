"""Shared training/validation engine — the L4 layer every entry point uses.

The reference copy-pasted this loop into six scripts; here it
is one implementation with the reference's latent bugs fixed:
- loss logging is interval-gated, no per-step barrier + all_reduce
  (reference did both every step purely for a log line, distributed.py:95-96)
- accumulated-loss logging averages the micro-step losses (reference logged
  only the last micro-batch, distributed_gradient_accumulation.py:91)
- scheduler.step() runs after the epoch's optimizer steps (reference stepped
  it first with the deprecated epoch arg, distributed.py:84)
"""
from __future__ import annotations

import time
from typing import Optional

import torch

from ..config import TrainConfig
from ..ops.batchnorm import (defer_num_batches_tracked,
                             flush_num_batches_tracked)
from . import dist as dist_utils
from .amp import DynamicLossScaler, autocast_ctx
from .metrics import AverageMeter, JsonlSink, ProgressMeter, accuracy
from .watchdog import StepWatchdog


def _grad_tensors(model):
    if getattr(model, "flat_grads", None) is not None:
        return [model.flat_grads]
    return [p.grad for p in model.parameters() if p.grad is not None]


def _zero_grads(model, optimizer):
    if hasattr(model, "zero_grad_buffer"):
        model.zero_grad_buffer()
    else:
        optimizer.zero_grad(set_to_none=False)


def train_one_epoch(model, loader, criterion, optimizer, epoch: int,
                    cfg: TrainConfig, device, scaler: Optional[DynamicLossScaler] = None,
                    sink: Optional[JsonlSink] = None, max_steps: Optional[int] = None) -> float:
    model.train()
    defer_num_batches_tracked(True)
    nprocs = dist_utils.get_world_size()
    accu = max(1, cfg.grad_accu_steps)
    losses = AverageMeter("Loss", ":.4e")
    step_time = AverageMeter("Time", ":6.3f")
    # true epoch-mean loss: accumulated on-device every step (one cheap add
    # kernel), materialised ONCE at epoch end — no per-step host sync and no
    # log_interval subsampling bias in the return value
    epoch_loss_sum = torch.zeros((), device=device)
    epoch_steps = 0
    watchdog = StepWatchdog(getattr(cfg, "stall_dump_s", None))
    end = time.time()

    for step, (images, labels) in enumerate(loader):
        if max_steps is not None and step >= max_steps:
            break
        watchdog.arm()
        images = images.to(device, non_blocking=True)
        labels = labels.to(device, non_blocking=True)
        if cfg.channels_last:
            images = images.to(memory_format=torch.channels_last)
        _zero_grads(model, optimizer)

        if accu == 1:
            with autocast_ctx(cfg.amp, device.type):
                outputs = model(images)
                loss = criterion(outputs, labels)
            backward_loss = scaler.scale_loss(loss) if scaler else loss
            backward_loss.backward()
            log_loss = loss.detach()
        else:
            # micro-batching: collectives elided on all but the final slice
            sub = max(1, images.size(0) // accu)
            log_loss = torch.zeros((), device=device)
            for a in range(accu):
                sub_images = images[a * sub:(a + 1) * sub]
                sub_labels = labels[a * sub:(a + 1) * sub]
                if sub_images.numel() == 0:
                    continue
                is_last = a == accu - 1
                ctx = model.no_sync() if (not is_last and hasattr(model, "no_sync")) \
                    else torch.enable_grad()
                with ctx:
                    with autocast_ctx(cfg.amp, device.type):
                        outputs = model(sub_images)
                        loss = criterion(outputs, sub_labels) / accu
                    backward_loss = scaler.scale_loss(loss) if scaler else loss
                    backward_loss.backward()
                log_loss += loss.detach()

        if hasattr(model, "finalize_backward"):
            model.finalize_backward()
        if scaler is not None:
            scaler.unscale_(_grad_tensors(model))
            scaler.step(optimizer)
        else:
            optimizer.step()
        flush_num_batches_tracked()
        watchdog.disarm()

        epoch_loss_sum += log_loss
        epoch_steps += 1
        step_time.update(time.time() - end)
        end = time.time()

        if step % cfg.log_interval == 0:
            # the ONLY host sync in the loop: materialise the loss at the
            # logging interval (a per-step float() would sync every step)
            reduced = dist_utils.reduce_mean(log_loss, nprocs)
            losses.update(float(reduced), images.size(0))
            if dist_utils.is_main_process():
                lr = optimizer.param_groups[0]["lr"]
                print(f"Epoch: [{epoch}][{step}/{len(loader)}]\t"
                      f"Loss {float(reduced):.4f}\tTime {step_time.val:.3f}\t"
                      f"LR {lr:.5f}", flush=True)
                if sink is not None:
                    sink.log(kind="train", epoch=epoch, step=step,
                             loss=float(reduced), lr=lr, step_time=step_time.val)
    # leave no deferral active outside the loop (direct bn_relu callers,
    # eval paths) and no pending bumps unapplied on early exits
    flush_num_batches_tracked()
    defer_num_batches_tracked(False)
    if epoch_steps == 0:
        return 0.0
    mean = dist_utils.reduce_mean(epoch_loss_sum / epoch_steps, nprocs)
    return float(mean)


@torch.no_grad()
def validate(model, loader, criterion, device, cfg: TrainConfig,
             sink: Optional[JsonlSink] = None, epoch: int = -1) -> float:
    """Distributed evaluation (reference utils/validation.py:7-52)."""
    model.eval()
    nprocs = dist_utils.get_world_size()
    batch_time = AverageMeter("Time", ":6.3f")
    losses = AverageMeter("Loss", ":.4e")
    top1 = AverageMeter("Acc@1", ":6.2f")
    top5 = AverageMeter("Acc@5", ":6.2f")
    progress = ProgressMeter(len(loader), [batch_time, losses, top1, top5],
                             prefix="Test: ")
    end = time.time()
    for i, (images, labels) in enumerate(loader):
        if cfg.max_eval_steps is not None and i >= cfg.max_eval_steps:
            break
        images = images.to(device, non_blocking=True)
        labels = labels.to(device, non_blocking=True)
        if cfg.channels_last:
            images = images.to(memory_format=torch.channels_last)
        with autocast_ctx(cfg.amp, device.type):
            outputs = model(images)
            loss = criterion(outputs, labels)
        acc1, acc5 = accuracy(outputs.float(), labels, topk=(1, 5))
        dist_utils.barrier()
        reduced_loss = dist_utils.reduce_mean(loss.detach().float(), nprocs)
        reduced_acc1 = dist_utils.reduce_mean(acc1, nprocs)
        reduced_acc5 = dist_utils.reduce_mean(acc5, nprocs)
        losses.update(float(reduced_loss), images.size(0))
        top1.update(float(reduced_acc1), images.size(0))
        top5.update(float(reduced_acc5), images.size(0))
        batch_time.update(time.time() - end)
        end = time.time()
        if i % cfg.log_interval == 0 and dist_utils.is_main_process():
            progress.display(i)
    if dist_utils.is_main_process():
        print(f" * Acc@1 {top1.avg:.3f} Acc@5 {top5.avg:.3f}", flush=True)
        if sink is not None:
            sink.log(kind="val", epoch=epoch, loss=losses.avg,
                     acc1=top1.avg, acc5=top5.avg)
    return top1.avg


def fit(model, train_loader, test_loader, train_sampler, criterion, optimizer,
        scheduler, cfg: TrainConfig, device,
        scaler: Optional[DynamicLossScaler] = None, start_epoch: int = 0) -> float:
    """Full training run: epochs, sampler reshuffle, eval, checkpoints."""
    from .checkpoint import save_checkpoint

    sink = JsonlSink(cfg.metrics_dir, enabled=dist_utils.is_main_process())
    best_acc = 0.0
    for epoch in range(start_epoch, cfg.epochs):
        if train_sampler is not None:
            train_sampler.set_epoch(epoch)
        t0 = time.time()
        train_one_epoch(model, train_loader, criterion, optimizer, epoch, cfg,
                        device, scaler=scaler, sink=sink,
                        max_steps=cfg.max_train_steps)
        scheduler.step()
        epoch_s = time.time() - t0
        if dist_utils.is_main_process():
            print(f"Epoch {epoch} done in {epoch_s:.2f}s", flush=True)
            if sink is not None:
                sink.log(kind="epoch", epoch=epoch, seconds=epoch_s)
        if cfg.eval_every_epoch:
            acc = validate(model, test_loader, criterion, device, cfg,
                           sink=sink, epoch=epoch)
            # >= so the first evaluation (possibly 0.0 on tiny synthetic
            # runs) still produces a best checkpoint
            if acc >= best_acc:
                best_acc = acc
                # keep the best checkpoint alongside the periodic ones
                save_checkpoint(cfg.ckpt_dir, cfg.arch, epoch, model,
                                optimizer, scheduler, scaler, best_acc,
                                tag="best")
        if cfg.save_epoch > 0 and (epoch + 1) % cfg.save_epoch == 0:
            save_checkpoint(cfg.ckpt_dir, cfg.arch, epoch, model, optimizer,
                            scheduler, scaler, best_acc)
    save_checkpoint(cfg.ckpt_dir, cfg.arch, cfg.epochs - 1, model, optimizer,
                    scheduler, scaler, best_acc, tag="final")
    return best_acc
