"""Native implicit-GEMM convolution (NHWC bf16, MFMA) with torch fallback.

MI355X-native replacement for the reference's cuDNN conv dependency
(every nn.Conv2d in reference utils/model.py — 3x3 s1/s2
and 1x1 convs of the CIFAR ResNet). The HIP kernels (csrc/conv_igemm.hip)
run when the tensor is a bf16 channels_last CUDA tensor; anything else
(CPU tests, fp32 mode, exotic shapes) runs torch's conv so the same module
is testable everywhere. ``MI355X_NATIVE_CONV=0`` forces the torch path for
parity A/B.
"""
from __future__ import annotations

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import _backend


def native_conv_wanted() -> bool:
    return os.environ.get("MI355X_NATIVE_CONV", "1") == "1"


def _native_ok(x: torch.Tensor, weight: torch.Tensor, stride, padding,
               dilation, groups) -> bool:
    if not native_conv_wanted() or not _backend.native_enabled(x):
        return False
    if x.dtype != torch.bfloat16 or weight.dtype != torch.bfloat16:
        return False
    if groups != 1 or dilation[0] != 1 or dilation[1] != 1:
        return False
    if stride[0] != stride[1] or padding[0] != padding[1]:
        return False
    # only take over when the tensors are already NHWC — in NCHW mode the
    # MIOpen path stays (no per-call layout conversions)
    if not x.is_contiguous(memory_format=torch.channels_last):
        return False
    # square filters only (the ResNet zoo is 3x3 / 1x1)
    return weight.shape[2] == weight.shape[3]


class _ConvIGEMM(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, stride: int, padding: int):
        x = x.contiguous(memory_format=torch.channels_last)
        weight = weight.contiguous(memory_format=torch.channels_last)
        ctx.save_for_backward(x, weight)
        ctx.stride, ctx.padding = stride, padding
        return _backend.C().conv_fwd_igemm(x, weight, stride, padding)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            # 180°-rotated, (C,R,S,K)-transposed filter for the dgrad GEMM
            wT = _backend.C().conv_build_wT(weight)
            dx = _backend.C().conv_dgrad_igemm(dy, wT, x.shape[2], x.shape[3],
                                               ctx.stride, ctx.padding)
        if ctx.needs_input_grad[1]:
            R = weight.shape[2]
            dw_f32 = _backend.C().conv_wgrad_igemm(dy, x, R, R,
                                                   ctx.stride, ctx.padding)
            # fp32 [K][R*S*C] is exactly the channels_last layout of (K,C,R,S)
            dw_view = dw_f32.view(weight.shape[0], R, R, weight.shape[1]) \
                            .permute(0, 3, 1, 2)
            dw = torch.empty_like(weight).copy_(dw_view)
        return dx, dw, None, None


def conv2d(x: torch.Tensor, weight: torch.Tensor, stride=(1, 1),
           padding=(0, 0), dilation=(1, 1), groups: int = 1) -> torch.Tensor:
    if _native_ok(x, weight, stride, padding, dilation, groups):
        return _ConvIGEMM.apply(x, weight, stride[0], padding[0])
    return F.conv2d(x, weight, None, stride, padding, dilation, groups)


class MI355Conv2d(nn.Conv2d):
    """nn.Conv2d whose hot path is the implicit-GEMM HIP kernel.

    Under bf16 autocast the inputs are cast explicitly (custom autograd
    Functions are invisible to autocast's casting); outside autocast the
    native path runs only when the tensors already are bf16.
    """

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        weight = self.weight
        if torch.is_autocast_enabled() and \
                torch.get_autocast_dtype("cuda") == torch.bfloat16 and x.is_cuda:
            x = x.to(torch.bfloat16)
            weight = weight.to(torch.bfloat16)
        return conv2d(x, weight, self.stride, self.padding, self.dilation,
                      self.groups)

    @classmethod
    def convert(cls, module: nn.Module) -> nn.Module:
        """Swap every plain nn.Conv2d (bias-free, as the ResNet zoo builds
        them) for MI355Conv2d, preserving parameters."""
        if type(module) is nn.Conv2d and module.bias is None:
            new = cls(module.in_channels, module.out_channels,
                      module.kernel_size, module.stride, module.padding,
                      module.dilation, module.groups, bias=False)
            new.weight = module.weight
            return new
        for name, child in module.named_children():
            setattr(module, name, cls.convert(child))
        return module
