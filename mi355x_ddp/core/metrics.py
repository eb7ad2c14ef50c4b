"""Metrics, meters and observability.

Console format matches the reference (utils/util.py:11-48) so output is
comparable side-by-side; additionally every logged record can go to a JSONL
sink (the reference's tensorboard_dir knob was dead code, utils/config.py:8).
"""
from __future__ import annotations

import json
import os
import time
from typing import Iterable, Optional, Sequence

import torch


class AverageMeter:
    """Running val/sum/count/avg (reference utils/util.py:11-32)."""

    def __init__(self, name: str, fmt: str = ":f"):
        self.name = name
        self.fmt = fmt
        self.reset()

    def reset(self):
        self.val = 0.0
        self.avg = 0.0
        self.sum = 0.0
        self.count = 0

    def update(self, val, n: int = 1):
        self.val = float(val)
        self.sum += float(val) * n
        self.count += n
        self.avg = self.sum / max(1, self.count)

    def __str__(self):
        fmtstr = "{name} {val" + self.fmt + "} ({avg" + self.fmt + "})"
        return fmtstr.format(**self.__dict__)


class ProgressMeter:
    """Formatted per-batch progress line (reference utils/util.py:34-48)."""

    def __init__(self, num_batches: int, meters: Sequence[AverageMeter], prefix: str = ""):
        self.batch_fmtstr = self._get_batch_fmtstr(num_batches)
        self.meters = meters
        self.prefix = prefix

    def display(self, batch: int):
        entries = [self.prefix + self.batch_fmtstr.format(batch)]
        entries += [str(meter) for meter in self.meters]
        print("\t".join(entries), flush=True)

    @staticmethod
    def _get_batch_fmtstr(num_batches: int) -> str:
        num_digits = len(str(num_batches // 1))
        fmt = "{:" + str(num_digits) + "d}"
        return "[" + fmt + "/" + fmt.format(num_batches) + "]"


@torch.no_grad()
def accuracy(output: torch.Tensor, target: torch.Tensor,
             topk: Iterable[int] = (1, 5)) -> list[torch.Tensor]:
    """Top-k accuracy in percent (reference utils/util.py:50-64).

    On GPU this runs the native class_rank kernel: one pass computing
    rank(target) per row — acc@k = mean(rank < k) — instead of ATen's
    sort-based topk + eq + k reductions.
    """
    from ..ops import _backend
    if _backend.native_enabled(output):
        rank = _backend.C().class_rank(output.contiguous(), target)
        return [(rank < k).float().mean().mul(100.0).reshape(1) for k in topk]
    maxk = max(topk)
    batch_size = target.size(0)
    _, pred = output.topk(maxk, 1, True, True)
    pred = pred.t()
    correct = pred.eq(target.view(1, -1).expand_as(pred))
    res = []
    for k in topk:
        correct_k = correct[:k].reshape(-1).float().sum(0, keepdim=True)
        res.append(correct_k.mul_(100.0 / batch_size))
    return res


class JsonlSink:
    """Append-only JSONL metrics sink; rank-0 only, cheap no-op elsewhere."""

    def __init__(self, directory: Optional[str], enabled: bool = True, name: str = "metrics"):
        self.path = None
        if enabled and directory:
            os.makedirs(directory, exist_ok=True)
            self.path = os.path.join(directory, f"{name}.jsonl")

    def log(self, **record):
        if self.path is None:
            return
        record.setdefault("ts", time.time())
        with open(self.path, "a") as f:
            f.write(json.dumps(record) + "\n")
