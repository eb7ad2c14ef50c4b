#!/usr/bin/env python3
"""Per-shape micro-benchmark: native implicit-GEMM conv vs MIOpen (F.conv2d).

Times fwd / dgrad / wgrad for every ResNet18 CIFAR conv shape at the bench
batch size, bf16 channels_last, and prints a table with speedups. Run on an
MI355X box; output goes to stdout (redirect into gpurun_out/).
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def timeit(fn, warmup=5, iters=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=256)
    args = p.parse_args()
    from mi355x_ddp import _C

    N = args.batch
    shapes = [
        ("stem 3x3", N, 3, 32, 32, 64, 3, 1, 1),
        ("s1 3x3", N, 64, 32, 32, 64, 3, 1, 1),
        ("s2 3x3/2", N, 64, 32, 32, 128, 3, 2, 1),
        ("s2 1x1/2", N, 64, 32, 32, 128, 1, 2, 0),
        ("s2 3x3", N, 128, 16, 16, 128, 3, 1, 1),
        ("s3 3x3/2", N, 128, 16, 16, 256, 3, 2, 1),
        ("s3 3x3", N, 256, 8, 8, 256, 3, 1, 1),
        ("s4 3x3/2", N, 256, 8, 8, 512, 3, 2, 1),
        ("s4 3x3", N, 512, 4, 4, 512, 3, 1, 1),
    ]
    print(f"{'shape':>10} {'pass':>6} {'MIOpen us':>10} {'igemm us':>9} "
          f"{'x':>6} {'TF/s':>7}")
    tot_m = tot_i = 0.0
    for name, n, c, h, w, k, r, stride, pad in shapes:
        x = torch.randn(n, c, h, w, device="cuda", dtype=torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        wt = (torch.randn(k, c, r, r, device="cuda", dtype=torch.bfloat16)
              / (c * r * r) ** 0.5).to(memory_format=torch.channels_last)
        pq = ((h + 2 * pad - r) // stride + 1)
        y = F.conv2d(x, wt, None, stride, pad)
        dy = torch.randn_like(y).to(memory_format=torch.channels_last)
        wT = wt.flip(2, 3).permute(1, 2, 3, 0).contiguous()
        flops = 2.0 * n * pq * pq * k * r * r * c

        # fwd
        tm = timeit(lambda: F.conv2d(x, wt, None, stride, pad))
        ti = timeit(lambda: _C.conv_fwd_igemm(x, wt, stride, pad))
        tot_m += tm; tot_i += ti
        print(f"{name:>10} {'fwd':>6} {tm:10.1f} {ti:9.1f} {tm/ti:6.2f} "
              f"{flops/ti/1e6:7.1f}")
        # dgrad (skip stem: input grad never needed there)
        if c != 3:
            xg = x.detach().requires_grad_(True)
            tm = timeit(lambda: torch.ops.aten.convolution_backward(
                dy, xg, wt, None, [stride, stride], [pad, pad], [1, 1], False,
                [0, 0], 1, [True, False, False]))
            ti = timeit(lambda: _C.conv_dgrad_igemm(dy, wT, h, w, stride, pad))
            tot_m += tm; tot_i += ti
            print(f"{name:>10} {'dgrad':>6} {tm:10.1f} {ti:9.1f} {tm/ti:6.2f} "
                  f"{flops/ti/1e6:7.1f}")
        # wgrad
        tm = timeit(lambda: torch.ops.aten.convolution_backward(
            dy, x, wt, None, [stride, stride], [pad, pad], [1, 1], False,
            [0, 0], 1, [False, True, False]))
        ti = timeit(lambda: _C.conv_wgrad_igemm(dy, x, r, r, stride, pad))
        tot_m += tm; tot_i += ti
        print(f"{name:>10} {'wgrad':>6} {tm:10.1f} {ti:9.1f} {tm/ti:6.2f} "
              f"{flops/ti/1e6:7.1f}")
    print(f"{'TOTAL':>10} {'':>6} {tot_m:10.1f} {tot_i:9.1f} "
          f"{tot_m/tot_i:6.2f}")


if __name__ == "__main__":
    main()
