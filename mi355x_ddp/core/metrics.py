"""Metrics, meters and observability.

Console output is byte-compatible with the reference's meter format
(utils/util.py:11-48) so runs are comparable side-by-side, but the
implementation here is its own: meters keep a single (total, n) pair and
derive everything, and every logged record can also go to a JSONL sink
(the reference's tensorboard_dir knob was dead code, utils/config.py:8).
"""
from __future__ import annotations

import json
import os
import time
from typing import Iterable, Optional, Sequence

import torch


class AverageMeter:
    """Running average with the reference's display format.

    State is one (total, n) accumulator pair plus the last value; ``avg``,
    ``sum`` and ``count`` are derived properties so the printed string
    ("{name} {val} ({avg})") matches reference utils/util.py output exactly.
    """

    __slots__ = ("name", "fmt", "val", "_total", "_n")

    def __init__(self, name: str, fmt: str = ":f"):
        self.name = name
        self.fmt = fmt
        self.reset()

    def reset(self) -> None:
        self.val = 0.0
        self._total = 0.0
        self._n = 0

    def update(self, val, n: int = 1) -> None:
        v = float(val)
        self.val = v
        self._total += v * n
        self._n += n

    @property
    def sum(self) -> float:
        return self._total

    @property
    def count(self) -> int:
        return self._n

    @property
    def avg(self) -> float:
        return self._total / self._n if self._n else 0.0

    def __str__(self) -> str:
        spec = self.fmt.lstrip(":")
        return f"{self.name} {self.val:{spec}} ({self.avg:{spec}})"


class ProgressMeter:
    """Per-batch progress line, reference-format ("prefix[ b/B]\\tmeters...")."""

    def __init__(self, num_batches: int, meters: Sequence[AverageMeter],
                 prefix: str = ""):
        width = len(str(num_batches))
        self._line_head = prefix + "[{:" + str(width) + "d}/" + str(num_batches) + "]"
        self.meters = meters
        self.prefix = prefix

    def display(self, batch: int) -> None:
        parts = [self._line_head.format(batch)]
        parts.extend(str(m) for m in self.meters)
        print("\t".join(parts), flush=True)


@torch.no_grad()
def accuracy(output: torch.Tensor, target: torch.Tensor,
             topk: Iterable[int] = (1, 5)) -> list[torch.Tensor]:
    """Top-k accuracy in percent (capability of reference utils/util.py:50-64).

    On GPU this runs the native class_rank kernel: one pass computing
    rank(target) per row — acc@k = mean(rank < k) — instead of ATen's
    sort-based topk + eq + k reductions. Tie handling: a logit equal to the
    target's counts as ranked above it only when its class index is smaller,
    which matches a stable descending sort (torch.topk's observed order).
    """
    from ..ops import _backend
    if _backend.native_enabled(output):
        rank = _backend.C().class_rank(output.contiguous(),
                                       target.contiguous())
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
