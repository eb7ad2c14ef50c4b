"""Model wrapping helpers for the three parallel modes.

- wrap_torch_ddp: torch's C++ reducer with bucket size tuned for xGMI (kept as
  an A/B reference against FlatDDP; reference wrapped at distributed.py:60).
- wrap_data_parallel: single-process multi-GPU nn.DataParallel semantics
  (reference dataparallel.py:47) — the low-priority compat path; DDP is the
  performance path (reference README.md:72-77 shows DP strictly slower).
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn


def wrap_torch_ddp(model: nn.Module, device_id: Optional[int],
                   bucket_cap_mb: Optional[float] = None,
                   static_graph: bool = False):
    from torch.nn.parallel import DistributedDataParallel as DDP
    kwargs = dict(bucket_cap_mb=25 if bucket_cap_mb is None else bucket_cap_mb,
                  gradient_as_bucket_view=True,
                  static_graph=static_graph)
    if device_id is not None:
        kwargs["device_ids"] = [device_id]
    return DDP(model, **kwargs)


def wrap_data_parallel(model: nn.Module, device_ids: List[int]):
    return nn.DataParallel(model, device_ids=device_ids,
                           output_device=device_ids[0])
