#!/usr/bin/env python3
"""RCCL collective sweep over xGMI — the data FlatDDP's bucket size should be
tuned against on an 8-GPU node. Launch with torchrun:

    torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \\
        tools/bench_collectives.py

Prints per-size all_reduce latency and algorithmic bandwidth from rank 0.
ResNet-18's whole gradient is 44.9 MB fp32 — the interesting region is
256 KB..64 MB (latency- to link-bound transition on 7x153 GB/s xGMI).
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local = int(os.environ.get("LOCAL_RANK", 0))
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    if torch.cuda.is_available():
        torch.cuda.set_device(local)
    dist.init_process_group(backend=backend)
    dev = torch.device("cuda", local) if torch.cuda.is_available() else "cpu"

    sizes = [4, 1024, 64 * 1024, 256 * 1024, 1 << 20, 4 << 20, 12 << 20,
             25 << 20, 45 << 20, 64 << 20]  # bytes
    if rank == 0:
        print(f"world={world}  backend={backend}  "
              f"NCCL_ALGO={os.environ.get('NCCL_ALGO', '<default>')}")
        print(f"{'dtype':>6} {'bytes':>12} {'us/call':>10} {'algbw GB/s':>11} "
              f"{'busbw GB/s':>11}")
    # fp32 = FlatDDP's default gradient buckets; bf16 = comm_dtype=bf16
    # shadow buckets (same element counts, half the wire bytes)
    for dtype, esz in ((torch.float32, 4), (torch.bfloat16, 2)):
        if dtype == torch.bfloat16 and backend == "gloo" and dev == "cpu":
            continue  # CPU-gloo bf16 all_reduce support varies; fp32 suffices
        for nbytes in sizes:
            t = torch.ones(nbytes // esz, device=dev, dtype=dtype)
            for _ in range(5):
                dist.all_reduce(t)
            if dev != "cpu":
                torch.cuda.synchronize()
            dist.barrier()
            iters = 20
            t0 = time.perf_counter()
            for _ in range(iters):
                dist.all_reduce(t)
            if dev != "cpu":
                torch.cuda.synchronize()
            el = (time.perf_counter() - t0) / iters
            if rank == 0:
                algbw = nbytes / el / 1e9
                busbw = algbw * 2 * (world - 1) / max(1, world)
                name = "fp32" if esz == 4 else "bf16"
                print(f"{name:>6} {nbytes:>12} {el * 1e6:>10.1f} "
                      f"{algbw:>11.2f} {busbw:>11.2f}", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
