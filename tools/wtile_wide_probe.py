#!/usr/bin/env python3
"""Agreement + timing probe for the wide-stage wgrad variants (wtile 8/9)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from mi355x_ddp import _C

def t(fn, it=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it * 1e6

for (n, c, h, w, k, r, stride, pad) in [
        (8, 128, 16, 16, 128, 3, 1, 1),
        (8, 64, 32, 32, 256, 1, 1, 0),
        (64, 512, 28, 28, 512, 3, 1, 1),
        (64, 256, 56, 56, 256, 3, 1, 1)]:
    x = torch.randn(n, c, h, w, device="cuda", dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    y = torch.randn(n, k, (h + 2 * pad - r) // stride + 1,
                    (w + 2 * pad - r) // stride + 1, device="cuda",
                    dtype=torch.bfloat16).to(memory_format=torch.channels_last)
    ref = _C.conv_wgrad_igemm(y, x, r, r, stride, pad, 1, 1).float()
    row = f"{c}->{k} {r}x{r}@{h}: "
    for wt in (2, 3, 8, 9):
        got = _C.conv_wgrad_igemm(y, x, r, r, stride, pad, 0, wt)
        rel = ((got.float() - ref).norm() / ref.norm()).item()
        us = t(lambda: _C.conv_wgrad_igemm(y, x, r, r, stride, pad, 0, wt))
        row += f" wt{wt}: rel={rel:.4f} {us:.0f}us |"
    mi = t(lambda: torch.ops.aten.convolution_backward(
        y, x, torch.empty(k, c, r, r, device="cuda", dtype=torch.bfloat16)
        .to(memory_format=torch.channels_last), None, [stride, stride],
        [pad, pad], [1, 1], False, [0, 0], 1, [False, True, False]))
    print(row + f" miopen {mi:.0f}us", flush=True)
print("probe ok")
