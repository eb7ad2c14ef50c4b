"""Mixed precision, apex-equivalent, no apex.

bf16-first on CDNA4: MFMA bf16 runs at ~16x the fp32 rate and needs no loss
scaling, so `amp="bf16"` is plain torch.autocast. `amp="fp16"` adds a dynamic
loss scaler with the apex behavior (scale loss, multi-tensor unscale +
inf/nan check as ONE HIP kernel, skip step on overflow, grow/backoff scale)
— reference used apex.amp at distributed_apex.py:86,119-120.
"""
from __future__ import annotations

import contextlib
from typing import List, Optional

import torch

from ..ops import _backend

# "bf16_o2" = apex-O2 equivalent: the MODEL's parameters are bf16 (fp32
# masters live in FusedSGD state), so autocast's per-forward weight casts
# disappear; activations run bf16 like plain "bf16"
_DTYPES = {"bf16": torch.bfloat16, "fp16": torch.float16,
           "bf16_o2": torch.bfloat16}


def autocast_ctx(mode: str, device_type: Optional[str] = None):
    """'fp32' -> nullcontext; 'bf16'/'fp16' -> torch.autocast."""
    if mode in (None, "fp32"):
        return contextlib.nullcontext()
    if mode not in _DTYPES:
        raise ValueError(f"unknown amp mode {mode!r}")
    device_type = device_type or ("cuda" if torch.cuda.is_available() else "cpu")
    return torch.autocast(device_type=device_type, dtype=_DTYPES[mode])


def multi_tensor_unscale_check(grads: List[torch.Tensor], inv_scale: float) -> bool:
    """grads *= inv_scale; returns True when any grad has inf/nan."""
    if not grads:
        return False
    if _backend.native_enabled(grads[0]):
        found = _backend.C().multi_tensor_unscale(grads, inv_scale)
        return bool(found.item())
    found = False
    for g in grads:
        g.mul_(inv_scale)
        if not found and not torch.isfinite(g).all():
            found = True
    return found


class DynamicLossScaler:
    """apex-style dynamic loss scaling (behavioral parity, native kernels)."""

    def __init__(self, init_scale: float = 2.0 ** 16, growth_factor: float = 2.0,
                 backoff_factor: float = 0.5, growth_interval: int = 2000):
        self.scale = init_scale
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self._growth_tracker = 0
        self.found_inf = False

    def scale_loss(self, loss: torch.Tensor) -> torch.Tensor:
        return loss * self.scale

    def unscale_(self, grads: List[torch.Tensor]) -> bool:
        self.found_inf = multi_tensor_unscale_check(grads, 1.0 / self.scale)
        return self.found_inf

    def step(self, optimizer: torch.optim.Optimizer) -> bool:
        """optimizer.step() unless the last unscale_ saw inf/nan; then update
        the scale. Returns True when the step ran."""
        stepped = False
        if not self.found_inf:
            optimizer.step()
            stepped = True
        self.update()
        return stepped

    def update(self):
        if self.found_inf:
            self.scale = max(self.scale * self.backoff_factor, 1.0)
            self._growth_tracker = 0
        else:
            self._growth_tracker += 1
            if self._growth_tracker >= self.growth_interval:
                self.scale *= self.growth_factor
                self._growth_tracker = 0
        self.found_inf = False

    def state_dict(self):
        return {"scale": self.scale, "growth_tracker": self._growth_tracker}

    def load_state_dict(self, sd):
        self.scale = sd["scale"]
        self._growth_tracker = sd["growth_tracker"]


def build_scaler(mode: str) -> Optional[DynamicLossScaler]:
    return DynamicLossScaler() if mode == "fp16" else None


def cast_model_bf16(model):
    """apex-O2 model prep: every parameter re-stored as bf16 (buffers —
    BN running stats — stay fp32). FusedSGD creates the fp32 masters at
    first step."""
    for p in model.parameters():
        p.data = p.data.to(torch.bfloat16)
    return model
