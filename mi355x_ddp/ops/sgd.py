"""Fused multi-tensor SGD with momentum + weight decay.

MI355X-native replacement for the per-parameter ATen SGD step the reference
runs (optimizer at distributed.py:63; SURVEY.md §2.2 N9 calls this the
BASELINE north-star): ONE HIP kernel launch updates every parameter of the
model — ResNet18 has ~62 tensors, so the eager step's ~180 kernel launches
collapse to 1. Math matches torch.optim.SGD exactly:

    m <- mu*m + g + wd*p      (momentum buffer, damping 0)
    p <- p - lr*m

bf16 parameters (the apex-O2-equivalent ``amp="bf16_o2"`` mode, where the
MODEL lives in bf16 and no per-step autocast weight casts exist) get an
fp32 MASTER copy in the optimizer state: the update runs in fp32 against
the master and the bf16 parameter is its rounded copy —

    m <- mu*m + g + wd*master ; master <- master - lr*m ; p <- bf16(master)

Mixed models (some fp32, some bf16 tensors) split into the two paths.
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
            p32, g32, m32 = [], [], []
            p16, g16, w16, m16 = [], [], [], []
            momentum = group["momentum"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if p.dtype == torch.bfloat16:
                    if "master" not in state:
                        state["master"] = p.detach().float()
                    if momentum != 0 and "momentum_buffer" not in state:
                        state["momentum_buffer"] = torch.zeros_like(
                            state["master"])
                    p16.append(p)
                    g16.append(p.grad)
                    w16.append(state["master"])
                    if momentum != 0:
                        m16.append(state["momentum_buffer"])
                else:
                    if momentum != 0 and "momentum_buffer" not in state:
                        state["momentum_buffer"] = torch.zeros_like(p)
                    p32.append(p)
                    g32.append(p.grad)
                    if momentum != 0:
                        m32.append(state["momentum_buffer"])
            if p32:
                fused_sgd_step(p32, g32, m32 if momentum != 0 else None,
                               lr=group["lr"], momentum=momentum,
                               weight_decay=group["weight_decay"])
            if p16:
                fused_sgd_o2_step(p16, g16, w16,
                                  m16 if momentum != 0 else None,
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


def fused_sgd_o2_step(params: List[torch.Tensor], grads: List[torch.Tensor],
                      masters: List[torch.Tensor],
                      momentum_bufs: Optional[List[torch.Tensor]],
                      lr: float, momentum: float, weight_decay: float) -> None:
    """bf16 params/grads, fp32 master + momentum (apex-O2 update)."""
    if params and _backend.native_enabled(params[0]):
        _backend.C().multi_tensor_sgd_o2(
            params, grads, masters,
            momentum_bufs if momentum_bufs is not None else [],
            lr, momentum, weight_decay)
        return
    for i, (p, g, w) in enumerate(zip(params, grads, masters)):
        gf = g.float()
        if weight_decay != 0:
            gf = gf.add(w, alpha=weight_decay)
        if momentum_bufs is not None:
            buf = momentum_bufs[i]
            buf.mul_(momentum).add_(gf)
            gf = buf
        w.add_(gf, alpha=-lr)
        p.copy_(w)
