"""Fused BatchNorm(+Add)(+ReLU) with native HIP kernels and a SyncBN mode.

MI355X-native replacement for the reference's dependency-provided kernels
(SURVEY.md §2.2 N2/N3/N4): on GPU the normalize/activation work is one fused
HIP kernel per tensor instead of ATen's separate BN and ReLU launches, and the
backward fuses the ReLU mask + BN reductions. SyncBatchNorm shares per-channel
{sum, sqsum, count} across ranks with ONE all_reduce in forward and one in
backward (reference used torch.nn.SyncBatchNorm, distributed.py:59).

The same autograd.Function drives:
  - CPU (pure torch math)  -> gloo-testable, including the cross-rank backward
  - GPU (HIP kernels)      -> the hardware path
so the distributed math is covered by CPU tests and numerics by GPU tests.
"""
from __future__ import annotations



import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from . import _backend


def _stats(x: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Per-channel (sum, sum-of-squares) in fp32 over N,H,W of an NCHW tensor."""
    if _backend.native_enabled(x):
        return _backend.C().bn_stats(x)
    xf = x.float()
    return xf.sum(dim=(0, 2, 3)), (xf * xf).sum(dim=(0, 2, 3))


def _fwd_apply(x, weight, bias, mean, invstd, relu, residual):
    if _backend.native_enabled(x):
        return _backend.C().bn_fwd(x, weight, bias, mean, invstd, relu,
                                   residual if residual is not None else torch.empty(0, device=x.device, dtype=x.dtype))
    shape = (1, -1, 1, 1)
    y = (x.float() - mean.view(shape)) * invstd.view(shape) * weight.float().view(shape) + bias.float().view(shape)
    y = y.to(x.dtype)
    if residual is not None:
        y = y + residual
    if relu:
        y = torch.relu(y)
    return y


def _bwd_reduce(dy, x, mean, invstd, y, relu):
    """Returns (sum_dy, sum_dy_xhat) per channel, with dy masked by relu(y)>0."""
    if _backend.native_enabled(x):
        return _backend.C().bn_bwd_reduce(dy, x, mean, invstd, y, relu)
    g = dy.float()
    if relu:
        g = g * (y > 0).float()
    xhat = (x.float() - mean.view(1, -1, 1, 1)) * invstd.view(1, -1, 1, 1)
    return g.sum(dim=(0, 2, 3)), (g * xhat).sum(dim=(0, 2, 3))


def _bwd_apply(dy, x, weight, mean, invstd, sum_dy, sum_dy_xhat, count, y, relu,
               training, need_dresidual):
    """dx (+ dresidual) for the fused BN(+add)(+relu)."""
    if _backend.native_enabled(x):
        out = _backend.C().bn_bwd(dy, x, weight.float().contiguous(), mean,
                                  invstd, sum_dy, sum_dy_xhat, float(count), y,
                                  relu, training, need_dresidual)
        return out[0], (out[1] if need_dresidual else None)
    g = dy.float()
    if relu:
        g = g * (y > 0).float()
    dresidual = g.to(x.dtype) if need_dresidual else None
    shape = (1, -1, 1, 1)
    w_is = (weight.float() * invstd).view(shape)
    if training:
        xhat = (x.float() - mean.view(shape)) * invstd.view(shape)
        dx = w_is * (g - sum_dy.view(shape) / count - xhat * sum_dy_xhat.view(shape) / count)
    else:
        dx = w_is * g
    return dx.to(x.dtype), dresidual


class _FusedBNFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, weight, bias, running_mean, running_var,
                training, momentum, eps, relu, process_group):
        # preserve channels_last end-to-end: NHWC tensors run the NHWC kernels,
        # no transposes are ever inserted around the fused op
        fmt = torch.channels_last if (x.dim() == 4 and x.is_contiguous(
            memory_format=torch.channels_last) and not x.is_contiguous()) \
            else torch.contiguous_format
        x = x.contiguous(memory_format=fmt)
        if residual is not None:
            residual = residual.contiguous(memory_format=fmt)
        ctx.fmt = fmt
        N, C, H, W = x.shape
        if training:
            # count is computed arithmetically (equal per-rank batches are
            # guaranteed by drop_last sharding) — no .item() host sync per BN
            # layer, which also keeps the op hipGraph-capturable.
            cnt = float(N * H * W)
            synced = process_group is not None and dist.is_initialized() \
                and dist.get_world_size(process_group) > 1
            if synced:
                cnt *= dist.get_world_size(process_group)
            if _backend.native_enabled(x):
                # packed {sum,sqsum} [2C]: one stats kernel, one optional RCCL
                # all_reduce, one finalize launch (mean/invstd + running-stats
                # update) — replaces the eager mean/var/rsqrt/lerp chain.
                packed = _backend.C().bn_stats_packed(x)
                if synced:
                    dist.all_reduce(packed, op=dist.ReduceOp.SUM,
                                    group=process_group)
                mean, invstd = _backend.C().bn_finalize(
                    packed, cnt, momentum, eps, running_mean, running_var)
            else:
                s, sq = _stats(x)
                if synced:
                    packed = torch.cat([s, sq])
                    dist.all_reduce(packed, op=dist.ReduceOp.SUM,
                                    group=process_group)
                    s, sq = packed[:C], packed[C:2 * C]
                mean = s / cnt
                var = sq / cnt - mean * mean
                var = var.clamp_min_(0.0)
                invstd = torch.rsqrt(var + eps)
                if running_mean is not None:
                    unbiased = var * (cnt / max(cnt - 1.0, 1.0))
                    running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                    running_var.mul_(1 - momentum).add_(unbiased, alpha=momentum)
        else:
            mean = running_mean.float()
            invstd = torch.rsqrt(running_var.float() + eps)
            cnt = float(N * H * W)
        y = _fwd_apply(x, weight, bias, mean, invstd, relu, residual)
        ctx.save_for_backward(x, weight, mean, invstd, y)
        ctx.relu = relu
        ctx.training = training
        ctx.count = cnt
        ctx.process_group = process_group
        ctx.has_residual = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, invstd, y = ctx.saved_tensors
        dy = dy.contiguous(memory_format=ctx.fmt)
        count = ctx.count
        C = x.shape[1]
        synced = ctx.training and ctx.process_group is not None \
            and dist.is_initialized() \
            and dist.get_world_size(ctx.process_group) > 1
        if _backend.native_enabled(x):
            packed = _backend.C().bn_bwd_reduce_packed(dy, x, mean, invstd, y,
                                                       ctx.relu)
        else:
            s_dy, s_dyx = _bwd_reduce(dy, x, mean, invstd, y, ctx.relu)
            packed = torch.cat([s_dy, s_dyx])
        # dgamma/dbeta are the LOCAL sums — the DP gradient all-reduce averages
        # them like every other parameter grad. The cross-rank-summed versions
        # are only for dx (whose formula needs the GLOBAL batch statistics).
        # copy=True: the all_reduce below mutates `packed` in place and these
        # must keep the LOCAL values
        dgamma = packed[C:].to(weight.dtype, copy=True)
        dbeta = packed[:C].to(weight.dtype, copy=True)
        if synced:
            dist.all_reduce(packed, op=dist.ReduceOp.SUM, group=ctx.process_group)
        sum_dy, sum_dy_xhat = packed[:C], packed[C:]
        dx, dresidual = _bwd_apply(dy, x, weight, mean, invstd, sum_dy,
                                   sum_dy_xhat, count, y, ctx.relu,
                                   ctx.training, ctx.has_residual)
        return (dx, dresidual, dgamma, dbeta, None, None, None, None, None, None, None)


def _group_for(bn: nn.Module):
    if isinstance(bn, MI355SyncBatchNorm):
        return bn.process_group if bn.process_group is not None else (
            dist.group.WORLD if dist.is_available() and dist.is_initialized() else None)
    return None


# Deferred num_batches_tracked updates: every BN otherwise launches a 4 us
# int64 add kernel per step just to bump its counter (~20 launches on
# ResNet18). With deferral on, forwards collect the counter buffers and
# flush_num_batches_tracked() bumps them all in ONE torch._foreach_add_
# per step. Modules with momentum=None (cumulative averaging READS the
# counter each forward) keep the eager update.
_NBT_DEFER = False
_NBT_PENDING = []


def defer_num_batches_tracked(enable: bool = True) -> None:
    global _NBT_DEFER
    _NBT_DEFER = enable


def flush_num_batches_tracked() -> None:
    # apply the deferred counter bumps (call once per optimizer step)
    if _NBT_PENDING:
        torch._foreach_add_(_NBT_PENDING, 1)
        _NBT_PENDING.clear()


def _run_fused(x, residual, bn: nn.Module, relu: bool):
    training = bn.training or not bn.track_running_stats
    if bn.training and bn.track_running_stats and bn.num_batches_tracked is not None:
        if _NBT_DEFER and bn.momentum is not None:
            _NBT_PENDING.append(bn.num_batches_tracked)
        else:
            bn.num_batches_tracked.add_(1)
    momentum = bn.momentum if bn.momentum is not None else (
        1.0 / float(bn.num_batches_tracked.item()) if bn.num_batches_tracked is not None else 0.1)
    return _FusedBNFunction.apply(
        x, residual, bn.weight, bn.bias, bn.running_mean, bn.running_var,
        training, momentum, bn.eps, relu, _group_for(bn))


def bn_relu(x: torch.Tensor, bn: nn.Module, relu: bool = True) -> torch.Tensor:
    """y = relu(bn(x)) as one fused op (module params/buffers come from `bn`)."""
    return _run_fused(x, None, bn, relu)


def bn_add_relu(x: torch.Tensor, residual: torch.Tensor, bn: nn.Module) -> torch.Tensor:
    """y = relu(bn(x) + residual) — the ResNet residual join, one kernel."""
    return _run_fused(x, residual, bn, True)


class MI355SyncBatchNorm(nn.BatchNorm2d):
    """Cross-rank BatchNorm (reference: torch SyncBatchNorm, distributed.py:59).

    Stats are shared with one packed all_reduce over RCCL/xGMI per forward and
    one per backward (torch's implementation all_gathers per rank). Works on
    CPU+gloo too so the math is testable without GPUs.
    """

    def __init__(self, *args, process_group=None, **kwargs):
        super().__init__(*args, **kwargs)
        self.process_group = process_group

    def forward(self, x):  # standalone use; fused paths call bn_relu directly
        return _run_fused(x, None, self, relu=False)

    @classmethod
    def convert_sync_batchnorm(cls, module: nn.Module, process_group=None) -> nn.Module:
        """Swap every BatchNorm2d for MI355SyncBatchNorm, preserving state."""
        out = module
        if isinstance(module, nn.BatchNorm2d) and not isinstance(module, cls):
            out = cls(module.num_features, module.eps, module.momentum,
                      module.affine, module.track_running_stats,
                      process_group=process_group)
            # preserve the source module's device/dtype: converting an
            # already-.cuda() model must not leave the new BN params on CPU
            ref = module.weight if module.affine else module.running_mean
            if ref is not None:
                out = out.to(device=ref.device, dtype=ref.dtype)
            if module.affine:
                with torch.no_grad():
                    out.weight.copy_(module.weight)
                    out.bias.copy_(module.bias)
            out.running_mean = module.running_mean
            out.running_var = module.running_var
            out.num_batches_tracked = module.num_batches_tracked
            out.training = module.training
        else:
            for name, child in module.named_children():
                setattr(out, name, cls.convert_sync_batchnorm(child, process_group))
        return out
