#!/usr/bin/env python3
"""Targeted wgrad PMC probe: run the big 3x3 shape's tile variants so a
rocprofv3 --pmc pass captures SQ_WAIT_ANY / SQ_LDS_BANK_CONFLICT per
variant (round-3 input: why is the 3x3 wgrad at ~45% of MIOpen's
bandwidth?)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from mi355x_ddp import _C

n, c, h, w, k, r, stride, pad = 64, 512, 28, 28, 512, 3, 1, 1
x = torch.randn(n, c, h, w, device="cuda", dtype=torch.bfloat16) \
    .to(memory_format=torch.channels_last)
wt = torch.randn(k, c, r, r, device="cuda", dtype=torch.bfloat16) \
    .to(memory_format=torch.channels_last)
dy = torch.randn(n, k, h, w, device="cuda", dtype=torch.bfloat16) \
    .to(memory_format=torch.channels_last)
for tile in (1, 2, 3):
    for _ in range(20):
        _C.conv_wgrad_igemm(dy, x, r, r, stride, pad, 0, tile)
    torch.cuda.synchronize()
print("probe done")
