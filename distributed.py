#!/usr/bin/env python3
"""DDP entry, launcher-driven (capability parity with reference distributed.py).

Launch (torchrun, preferred):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 distributed.py --batch_size 256
Legacy torch.distributed.launch --local_rank injection is also accepted
(reference distributed.py:19,27-31).
"""
import argparse
import os

import torch

from mi355x_ddp.config import add_common_args, config_from_args
from mi355x_ddp.core import dist as dist_utils
from mi355x_ddp.core.worker import main_worker


def main():
    p = argparse.ArgumentParser(description="MI355X DDP training (launcher-driven)")
    p.add_argument("--local_rank", "--local-rank", default=None, type=int,
                   help="injected by torch.distributed.launch (legacy)")
    add_common_args(p)
    args = p.parse_args()
    cfg = config_from_args(args)

    if "RANK" in os.environ:  # torchrun path
        rank, local_rank, world_size = dist_utils.init_from_env(cfg.backend)
        main_worker(local_rank, world_size, cfg, rank=rank, init_pg=False)
    else:  # legacy launcher / single process
        local_rank = args.local_rank if args.local_rank is not None else 0
        world_size = torch.cuda.device_count() or 1
        main_worker(local_rank, world_size, cfg)


if __name__ == "__main__":
    main()
