#!/usr/bin/env python3
"""DDP + mixed-precision entry (capability parity with reference distributed_apex.py,
WITHOUT apex): CDNA4-native bf16 autocast by default — bf16 MFMA needs no loss
scaler — and `--amp fp16` for apex-behavioral parity (dynamic loss scaling with
a HIP multi-tensor unscale/inf-check kernel; reference used apex.amp at
distributed_apex.py:86,119-120).

Run: python distributed_apex.py --batch_size 256   # bf16 (apex O1 equivalent)
     python distributed_apex.py --amp fp16         # apex-parity loss scaling
     python distributed_apex.py --amp bf16_o2      # apex O2: bf16 model +
                                                   # fp32 master weights
"""
import argparse

import torch
import torch.multiprocessing as mp

from mi355x_ddp.config import add_common_args, config_from_args
from mi355x_ddp.core.worker import main_worker


def main():
    p = argparse.ArgumentParser(description="MI355X DDP + AMP training")
    p.add_argument("--nprocs", default=None, type=int)
    add_common_args(p)
    args = p.parse_args()
    cfg = config_from_args(args)
    if cfg.amp == "fp32":  # default for this entry is mixed precision
        cfg = cfg.replace(amp="bf16")
    if getattr(args, "channels_last", None) is None and             torch.cuda.is_available():
        # the AMP entry defaults to the fast path: NHWC + native igemm conv
        cfg = cfg.replace(channels_last=True)
    nprocs = args.nprocs or torch.cuda.device_count() or 1
    if nprocs == 1:
        main_worker(0, 1, cfg)
    else:
        mp.spawn(main_worker, nprocs=nprocs, args=(nprocs, cfg))


if __name__ == "__main__":
    main()
