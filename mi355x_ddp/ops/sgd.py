"""Fused multi-tensor SGD with momentum + weight decay.

MI355X-native replacement for the per-parameter ATen SGD step the reference
runs (optimizer at distributed.py:63; SURVEY.md §2.2 N9 calls this the
BASELINE north-star): ONE HIP kernel launch updates every parameter of the
model — ResNet18 has ~62 tensors, so the eager step's ~180 kernel launches
collapse to 1. Math matches torch.optim.SGD exactly:

    m <- mu*m + g + wd*p      (momentum buffer, damping 0)
    p <- p - lr*m

Supports a flat-parameter fast path (one contiguous buffer, used by
parallel.FlatDDP) and the generic multi-tensor path.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from . import _backend


class FusedSGD(torch.optim.Optimizer):
    def __init__(self, params, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            params: List[torch.Tensor] = []
            grads: List[torch.Tensor] = []
            bufs: List[torch.Tensor] = []
            momentum = group["momentum"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                params.append(p)
                grads.append(p.grad)
                state = self.state[p]
                if momentum != 0:
                    if "momentum_buffer" not in state:
                        state["momentum_buffer"] = torch.zeros_like(p)
                    bufs.append(state["momentum_buffer"])
            if not params:
                continue
            fused_sgd_step(params, grads, bufs if momentum != 0 else None,
                           lr=group["lr"], momentum=momentum,
                           weight_decay=group["weight_decay"])
        return loss


def fused_sgd_step(params: List[torch.Tensor], grads: List[torch.Tensor],
                   momentum_bufs: Optional[List[torch.Tensor]],
                   lr: float, momentum: float, weight_decay: float) -> None:
    if params and _backend.native_enabled(params[0]):
        _backend.C().multi_tensor_sgd(params, grads,
                                      momentum_bufs if momentum_bufs is not None else [],
                                      lr, momentum, weight_decay)
        return
    # torch fallback, same math (used on CPU and for parity tests)
    for i, (p, g) in enumerate(zip(params, grads)):
        if weight_decay != 0:
            g = g.add(p, alpha=weight_decay)
        if momentum_bufs is not None:
            buf = momentum_bufs[i]
            buf.mul_(momentum).add_(g)
            g = buf
        p.add_(g, alpha=-lr)
