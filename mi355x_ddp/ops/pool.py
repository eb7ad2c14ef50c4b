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


class _MaxPoolNHWC(torch.autograd.Function):
    """NHWC max pool with recorded argmax; backward is a GATHER (each input
    pixel sums the dy of the overlapping windows that chose it — no atomics).
    First-max tie-break matches ATen."""

    @staticmethod
    def forward(ctx, x, k: int, stride: int, pad: int):
        y, idx = _backend.C().maxpool_fwd(x, k, stride, pad)
        ctx.save_for_backward(idx)
        ctx.meta = (x.shape[2], x.shape[3], k, stride, pad)
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        H, W, k, stride, pad = ctx.meta
        dx = _backend.C().maxpool_bwd(
            dy.contiguous(memory_format=torch.channels_last), idx,
            H, W, k, stride, pad)
        return dx, None, None, None


class MI355MaxPool2d(nn.MaxPool2d):
    """nn.MaxPool2d whose GPU NHWC path is the native HIP kernel (the
    ImageNet stem's 3x3 s2 pool); everything else falls back to ATen."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        k = self.kernel_size if isinstance(self.kernel_size, int) \
            else self.kernel_size[0]
        s = self.stride if isinstance(self.stride, int) else self.stride[0]
        p = self.padding if isinstance(self.padding, int) else self.padding[0]
        if (_backend.native_enabled(x) and not self.ceil_mode and
                self.dilation == 1 and
                x.is_contiguous(memory_format=torch.channels_last) and
                x.shape[1] % 8 == 0 and x.dtype in
                (torch.bfloat16, torch.float16)):
            return _MaxPoolNHWC.apply(x, k, s, p)
        return super().forward(x)

    @classmethod
    def convert(cls, module: nn.Module) -> nn.Module:
        if type(module) is nn.MaxPool2d:
            return cls(module.kernel_size, module.stride, module.padding,
                       module.dilation, ceil_mode=module.ceil_mode)
        for name, child in module.named_children():
            setattr(module, name, cls.convert(child))
        return module
