#!/usr/bin/env python3
"""DDP + gradient accumulation entry (capability parity with reference
distributed_gradient_accumulation.py): the per-GPU batch is sliced into
--grad_accu_steps micro-batches; all but the last run under no_sync() so the
RCCL all-reduce fires once per outer batch (reference :90-111). The logged
loss is the correctly averaged accumulated loss (reference bug at :91 fixed).

Run: python distributed_gradient_accumulation.py --grad_accu_steps 4
"""
import argparse

import torch
import torch.multiprocessing as mp

from mi355x_ddp.config import add_common_args, config_from_args
from mi355x_ddp.core.worker import main_worker


def main():
    p = argparse.ArgumentParser(description="MI355X DDP + gradient accumulation")
    p.add_argument("--grad_accu_steps", default=4, type=int,
                   help="micro-steps per optimizer step")
    p.add_argument("--nprocs", default=None, type=int)
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
