"""Global average pooling with a native HIP kernel (reference utils/model.py:76
used AdaptiveAvgPool2d((1,1)) -> ATen CUDA kernels)."""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import _backend


class _GlobalAvgPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        return _backend.C().gap_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        return _backend.C().gap_bwd(dy.contiguous(), x)


class GlobalAvgPool2d(nn.Module):
    """AdaptiveAvgPool2d((1,1)) equivalent; native kernel on GPU (one fused
    reduction per (n,c), NHWC-coalesced), eager mean on CPU."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if _backend.native_enabled(x):
            return _GlobalAvgPool.apply(x)
        return F.adaptive_avg_pool2d(x, (1, 1))
