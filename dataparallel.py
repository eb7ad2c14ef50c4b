#!/usr/bin/env python3
"""Single-process DataParallel entry (capability parity with reference
dataparallel.py). Kept for CLI-surface completeness — DDP is the performance
path (reference's own table shows DP strictly slower, README.md:72-77). The
reference's unshuffled-train-loader bug (dataparallel.py:53) is fixed.

Run: python dataparallel.py --gpu 0,1,2,3 --batch_size 256
"""
import argparse
import os


def main():
    p = argparse.ArgumentParser(description="MI355X DataParallel training")
    p.add_argument("--gpu", default=None, type=str,
                   help="comma-separated device ids, e.g. '0,1,2,3'")
    from mi355x_ddp.config import add_common_args
    add_common_args(p)
    args = p.parse_args()
    # Set visibility BEFORE torch initialises HIP (reference ordering quirk at
    # dataparallel.py:27-29 noted in SURVEY §2.1).
    if args.gpu:
        os.environ["CUDA_VISIBLE_DEVICES"] = args.gpu
    run(args)


def run(args, amp_default="fp32"):
    import torch
    from mi355x_ddp.config import config_from_args
    from mi355x_ddp.core.engine import fit
    from mi355x_ddp.core.worker import build_training, init_seeds
    from mi355x_ddp.data import build_loaders
    from mi355x_ddp.parallel import wrap_data_parallel

    cfg = config_from_args(args)
    if cfg.amp == "fp32" and amp_default != "fp32":
        cfg = cfg.replace(amp=amp_default)
    init_seeds(cfg.seed)
    device = torch.device("cuda", 0) if torch.cuda.is_available() else torch.device("cpu")
    model, criterion, optimizer, scheduler, scaler = build_training(
        cfg, device, world_size=1, rank=0, distributed=False, wrap="none")
    if torch.cuda.is_available() and torch.cuda.device_count() > 1:
        model = wrap_data_parallel(model, list(range(torch.cuda.device_count())))
    train_loader, test_loader, _ = build_loaders(cfg, 1, 0, distributed=False)
    fit(model, train_loader, test_loader, None, criterion, optimizer,
        scheduler, cfg, device, scaler=scaler)


if __name__ == "__main__":
    main()
