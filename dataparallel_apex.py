#!/usr/bin/env python3
"""DataParallel + mixed precision entry (capability parity with reference
dataparallel_apex.py, without apex): bf16 autocast by default, `--amp fp16`
for apex-style dynamic loss scaling.

Run: python dataparallel_apex.py --gpu 0,1,2,3 --batch_size 256
"""
import argparse
import os

from dataparallel import run


def main():
    p = argparse.ArgumentParser(description="MI355X DataParallel + AMP training")
    p.add_argument("--gpu", default=None, type=str)
    from mi355x_ddp.config import add_common_args
    add_common_args(p)
    args = p.parse_args()
    if args.gpu:
        os.environ["CUDA_VISIBLE_DEVICES"] = args.gpu
    run(args, amp_default="bf16")


if __name__ == "__main__":
    main()
