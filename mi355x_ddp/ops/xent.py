"""Fused softmax + cross-entropy (reference: nn.CrossEntropyLoss at distributed.py:61).

One HIP kernel computes the row max, log-sum-exp and NLL in a single pass
(ATen launches softmax + nll separately); backward is one kernel producing
softmax(x) - onehot scaled by 1/N. CPU fallback is the same math in torch.
"""
from __future__ import annotations

import torch

from . import _backend


class _SoftmaxXentFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target):
        logits = logits.contiguous()
        if _backend.native_enabled(logits):
            loss, lse = _backend.C().xent_fwd(logits, target)
        else:
            lf = logits.float()
            lse = torch.logsumexp(lf, dim=1)
            loss = (lse - lf.gather(1, target.view(-1, 1)).squeeze(1)).mean()
        ctx.save_for_backward(logits, target, lse)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, target, lse = ctx.saved_tensors
        if _backend.native_enabled(logits):
            dx = _backend.C().xent_bwd(logits, target, lse, dloss)
        else:
            lf = logits.float()
            p = torch.exp(lf - lse.unsqueeze(1))
            p.scatter_add_(1, target.view(-1, 1),
                           torch.full_like(target.view(-1, 1), -1.0, dtype=p.dtype))
            dx = (p * (dloss.float() / logits.shape[0])).to(logits.dtype)
        return dx, None


def softmax_cross_entropy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Mean-reduced cross entropy over a (N, classes) logits tensor."""
    return _SoftmaxXentFunction.apply(logits, target)


class SoftmaxCrossEntropy(torch.nn.Module):
    """Drop-in for nn.CrossEntropyLoss(reduction='mean') on 2D logits."""

    def forward(self, logits, target):
        return softmax_cross_entropy(logits, target)
