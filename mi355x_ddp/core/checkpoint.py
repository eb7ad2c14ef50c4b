"""Rank-0 checkpointing (reference documented the pattern but shipped no code:
tutorials/2.:89-95 'save only on rank 0', dead save_epoch knob utils/config.py:7).

Layout: ckpts/{arch}_epoch{E}.pt and ckpts/{arch}_last.pt containing
{epoch, model (unwrapped module state), optimizer, scheduler, scaler, best_acc}.
"""
from __future__ import annotations

import os
from typing import Optional

import torch

from . import dist as dist_utils


def _unwrap(model):
    return model.module if hasattr(model, "module") else model


def save_checkpoint(ckpt_dir: str, arch: str, epoch: int, model, optimizer=None,
                    scheduler=None, scaler=None, best_acc: float = 0.0,
                    tag: Optional[str] = None) -> Optional[str]:
    if not dist_utils.is_main_process():
        return None
    os.makedirs(ckpt_dir, exist_ok=True)
    state = {
        "epoch": epoch,
        "arch": arch,
        "model": _unwrap(model).state_dict(),
        "optimizer": optimizer.state_dict() if optimizer is not None else None,
        "scheduler": scheduler.state_dict() if scheduler is not None else None,
        "scaler": scaler.state_dict() if scaler is not None else None,
        "best_acc": best_acc,
    }
    path = os.path.join(ckpt_dir, f"{arch}_{tag or f'epoch{epoch}'}.pt")
    torch.save(state, path)
    # {arch}_last.pt is a hardlink to the tagged file — one serialisation
    # pass, not two (ResNet50+ states are hundreds of MB)
    last = os.path.join(ckpt_dir, f"{arch}_last.pt")
    try:
        if os.path.lexists(last):
            os.remove(last)
        os.link(path, last)
    except OSError:  # cross-device / FS without hardlinks
        import shutil
        shutil.copyfile(path, last)
    return path


def load_checkpoint(path: str, model, optimizer=None, scheduler=None,
                    scaler=None, map_location="cpu") -> dict:
    state = torch.load(path, map_location=map_location, weights_only=False)
    _unwrap(model).load_state_dict(state["model"])
    if optimizer is not None and state.get("optimizer"):
        optimizer.load_state_dict(state["optimizer"])
    if scheduler is not None and state.get("scheduler"):
        scheduler.load_state_dict(state["scheduler"])
    if scaler is not None and state.get("scaler"):
        scaler.load_state_dict(state["scaler"])
    return state
