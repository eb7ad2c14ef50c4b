"""Process-group runtime: rendezvous, collectives helpers, rank utilities.

MI355X equivalent of the reference's L1 layer: on ROCm the
torch.distributed "nccl" backend IS RCCL, and single-node traffic runs over the
7 point-to-point xGMI links per GPU. Rendezvous is a TCP store on 127.0.0.1 by
default (reference used ``tcp://ip:port``, distributed.py:45-50); torchrun-style
env-var init is also supported.
"""
from __future__ import annotations

import contextlib
import datetime
import os
from typing import Dict, Optional

import torch
import torch.distributed as dist


def backend_for_device() -> str:
    return "nccl" if torch.cuda.is_available() else "gloo"


def init_distributed(rank: int, world_size: int, ip: str = "127.0.0.1",
                     port: int = 23456, backend: Optional[str] = None,
                     device_id: Optional[int] = None,
                     timeout_s: int = 600) -> None:
    """Create the process group (reference: dist.init_process_group at distributed.py:49).

    Works for both launch modes: torchrun/env (MASTER_ADDR set by launcher) and
    explicit tcp:// rendezvous for mp.spawn children.
    """
    if dist.is_initialized():
        return
    backend = backend or backend_for_device()
    if device_id is not None and torch.cuda.is_available():
        torch.cuda.set_device(device_id)
    init_method = f"tcp://{ip}:{port}"
    dist.init_process_group(
        backend=backend,
        init_method=init_method,
        world_size=world_size,
        rank=rank,
        timeout=datetime.timedelta(seconds=timeout_s),
    )


def init_from_env(backend: Optional[str] = None, timeout_s: int = 600) -> tuple[int, int, int]:
    """torchrun path: RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from env.

    Returns (rank, local_rank, world_size). Falls back to single-process when
    no launcher env is present.
    """
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    world_size = int(os.environ.get("WORLD_SIZE", 1))
    if dist.is_initialized():
        return rank, local_rank, world_size
    backend = backend or backend_for_device()
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    if "MASTER_ADDR" in os.environ:
        dist.init_process_group(backend=backend,
                                timeout=datetime.timedelta(seconds=timeout_s))
    else:
        dist.init_process_group(
            backend=backend, init_method="tcp://127.0.0.1:29512",
            world_size=1, rank=0,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    return rank, local_rank, world_size


def is_dist() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_dist() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_dist() else 1


def is_main_process() -> bool:
    return get_rank() == 0


def barrier() -> None:
    if is_dist():
        dist.barrier()


def reduce_mean(tensor: torch.Tensor, nprocs: Optional[int] = None) -> torch.Tensor:
    """Average a tensor across ranks (reference utils/util.py:5-9).

    Clone-then-all_reduce(SUM)/world; returns the input unchanged when not
    distributed so callers never branch.
    """
    if not is_dist():
        return tensor
    nprocs = nprocs or get_world_size()
    rt = tensor.detach().clone()
    dist.all_reduce(rt, op=dist.ReduceOp.SUM)
    rt /= nprocs
    return rt


def cleanup() -> None:
    if is_dist():
        dist.destroy_process_group()


@contextlib.contextmanager
def count_collectives() -> "Dict[str, int]":
    """Debug aux (collective launch-count assertions).

    Patches the torch.distributed collectives and yields a dict of call
    counts, so tests can assert communication BEHAVIOR (e.g. no_sync elides
    every gradient all-reduce; SyncBN issues exactly one all_reduce per
    forward) by count rather than only by value.
    """
    counts: Dict[str, int] = {}
    originals = {}
    for name in ("all_reduce", "all_gather", "broadcast", "barrier",
                 "reduce_scatter_tensor", "all_gather_into_tensor"):
        orig = getattr(dist, name)
        originals[name] = orig

        def make(name, orig):
            def wrapper(*a, **kw):
                counts[name] = counts.get(name, 0) + 1
                return orig(*a, **kw)
            return wrapper
        setattr(dist, name, make(name, orig))
    try:
        yield counts
    finally:
        for name, orig in originals.items():
            setattr(dist, name, orig)
