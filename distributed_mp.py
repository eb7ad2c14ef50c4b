#!/usr/bin/env python3
"""DDP entry, self-spawning (capability parity with reference distributed_mp.py).

One process per visible GPU via mp.spawn (reference distributed_mp.py:41-45);
per-rank seeding happens inside main_worker (reference init_seeds :29-39).
Run: python distributed_mp.py --batch_size 256 --epochs 200
"""
import argparse

import torch
import torch.multiprocessing as mp

from mi355x_ddp.config import add_common_args, config_from_args
from mi355x_ddp.core.worker import main_worker


def main():
    p = argparse.ArgumentParser(description="MI355X DDP training (mp.spawn)")
    p.add_argument("--nprocs", default=None, type=int,
                   help="process count; default = visible GPU count")
    add_common_args(p)
    args = p.parse_args()
    cfg = config_from_args(args)
    nprocs = args.nprocs or torch.cuda.device_count() or 1
    if nprocs == 1:
        main_worker(0, 1, cfg)
    else:
        mp.spawn(main_worker, nprocs=nprocs, args=(nprocs, cfg))


if __name__ == "__main__":
    main()
