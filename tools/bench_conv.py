#!/usr/bin/env python3
"""Per-shape micro-benchmark: native implicit-GEMM conv vs MIOpen (F.conv2d).

Times fwd / dgrad / wgrad for every conv shape of a model config at the bench
batch size, bf16 channels_last, and prints a table with speedups. Run on an
MI355X box; output goes to stdout (redirect into gpurun_out/).

Default: the ResNet18 CIFAR table (round-1 baseline format). --arch/--image-
size collects the shape set from the actual model with forward hooks, e.g.
`--arch resnet50 --image-size 224 --batch 64` for BASELINE config 5.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn
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


R18_CIFAR_SHAPES = [
    # (name, C, H, W, K, R, stride, pad)
    ("stem 3x3", 3, 32, 32, 64, 3, 1, 1),
    ("s1 3x3", 64, 32, 32, 64, 3, 1, 1),
    ("s2 3x3/2", 64, 32, 32, 128, 3, 2, 1),
    ("s2 1x1/2", 64, 32, 32, 128, 1, 2, 0),
    ("s2 3x3", 128, 16, 16, 128, 3, 1, 1),
    ("s3 3x3/2", 128, 16, 16, 256, 3, 2, 1),
    ("s3 3x3", 256, 8, 8, 256, 3, 1, 1),
    ("s4 3x3/2", 256, 8, 8, 512, 3, 2, 1),
    ("s4 3x3", 512, 4, 4, 512, 3, 1, 1),
]


def collect_model_shapes(arch: str, image_size: int):
    """Unique conv shapes (with multiplicities) of a model at this input
    size, discovered by a CPU forward pass with hooks."""
    from mi355x_ddp.models import build_model
    model = build_model(arch)
    shapes = {}  # (C,H,W,K,R,stride,pad) -> count

    def hook(mod, inp, out):
        x = inp[0]
        key = (x.shape[1], x.shape[2], x.shape[3], mod.out_channels,
               mod.kernel_size[0], mod.stride[0], mod.padding[0])
        shapes[key] = shapes.get(key, 0) + 1

    handles = [m.register_forward_hook(hook) for m in model.modules()
               if isinstance(m, nn.Conv2d)]
    with torch.no_grad():
        model(torch.zeros(1, 3, image_size, image_size))
    for h in handles:
        h.remove()
    out = []
    for (c, h, w, k, r, s, p), cnt in shapes.items():
        tag = f"{r}x{r}" + (f"/{s}" if s > 1 else "")
        out.append((f"{c}->{k} {tag}@{h}", c, h, w, k, r, s, p, cnt))
    return out


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=256)
    p.add_argument("--arch", type=str, default=None,
                   help="collect shapes from this model instead of the "
                        "static ResNet18 CIFAR table")
    p.add_argument("--image-size", type=int, default=32)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    args = p.parse_args()
    from mi355x_ddp import _C

    N = args.batch
    if args.arch:
        rows = collect_model_shapes(args.arch, args.image_size)
    else:
        rows = [(name, c, h, w, k, r, s, p, 1)
                for (name, c, h, w, k, r, s, p) in R18_CIFAR_SHAPES]

    print(f"{'shape':>18} {'pass':>6} {'n':>3} {'MIOpen us':>10} "
          f"{'igemm us':>9} {'best us':>9} {'cfg':>9} {'x':>6} {'TF/s':>7}")
    tot_m = tot_i = tot_b = 0.0  # weighted by multiplicity
    for name, c, h, w, k, r, stride, pad, cnt in rows:
        x = torch.randn(N, c, h, w, device="cuda", dtype=torch.bfloat16) \
            .to(memory_format=torch.channels_last)
        wt = (torch.randn(k, c, r, r, device="cuda", dtype=torch.bfloat16)
              / (c * r * r) ** 0.5).to(memory_format=torch.channels_last)
        ph = ((h + 2 * pad - r) // stride + 1)
        pw = ((w + 2 * pad - r) // stride + 1)
        y = F.conv2d(x, wt, None, stride, pad)
        dy = torch.randn_like(y).to(memory_format=torch.channels_last)
        wT = wt.flip(2, 3).permute(1, 2, 3, 0).contiguous()
        flops = 2.0 * N * ph * pw * k * r * r * c

        def row(pss, tm, ti, tb, cfg):
            nonlocal tot_m, tot_i, tot_b
            tot_m += tm * cnt
            tot_i += ti * cnt
            tot_b += tb * cnt
            print(f"{name:>18} {pss:>6} {cnt:>3} {tm:10.1f} {ti:9.1f} "
                  f"{tb:9.1f} {cfg:>9} {tm/tb:6.2f} {flops/tb/1e6:7.1f}",
                  flush=True)

        def sweep(cands, run):
            best, bcfg = float("inf"), None
            for cand in cands:
                t = timeit(lambda: run(cand), warmup=2, iters=args.iters)
                if t < best:
                    best, bcfg = t, cand
            return best, str(bcfg)

        kw = dict(warmup=args.warmup, iters=args.iters)
        # fwd
        tm = timeit(lambda: F.conv2d(x, wt, None, stride, pad), **kw)
        ti = timeit(lambda: _C.conv_fwd_igemm(x, wt, stride, pad), **kw)
        tb, cfg = sweep((64, 128, 228),
                        lambda t: _C.conv_fwd_igemm(x, wt, stride, pad, t))
        row("fwd", tm, ti, tb, cfg)
        # dgrad (skip stem: input grad never needed there)
        if c != 3:
            xg = x.detach().requires_grad_(True)
            tm = timeit(lambda: torch.ops.aten.convolution_backward(
                dy, xg, wt, None, [stride, stride], [pad, pad], [1, 1], False,
                [0, 0], 1, [True, False, False]), **kw)
            ti = timeit(lambda: _C.conv_dgrad_igemm(dy, wT, h, w, stride, pad),
                        **kw)
            tb, cfg = sweep((64, 128, 228), lambda t: _C.conv_dgrad_igemm(
                dy, wT, h, w, stride, pad, t))
            row("dgrad", tm, ti, tb, cfg)
        # wgrad: sweep tile variants x split factors
        tm = timeit(lambda: torch.ops.aten.convolution_backward(
            dy, x, wt, None, [stride, stride], [pad, pad], [1, 1], False,
            [0, 0], 1, [False, True, False]), **kw)
        ti = timeit(lambda: _C.conv_wgrad_igemm(dy, x, r, r, stride, pad), **kw)
        wcands = [(t, s) for t in (1, 2, 3, 4, 5, 6)
                  for s in (0, 1, 32, 128)] \
            if c != 3 else [(t, s) for t in (1, 4)
                            for s in (0, 32, 128, 512)]
        tb, cfg = sweep(wcands, lambda ts: _C.conv_wgrad_igemm(
            dy, x, r, r, stride, pad, ts[1], ts[0]))
        row("wgrad", tm, ti, tb, cfg)
        del x, wt, y, dy, wT
    print(f"{'TOTAL':>18} {'':>6} {'':>3} {tot_m:10.1f} {tot_i:9.1f} "
          f"{tot_b:9.1f} {'':>9} {tot_m/tot_b:6.2f}")


if __name__ == "__main__":
    main()
