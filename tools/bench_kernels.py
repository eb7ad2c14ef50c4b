#!/usr/bin/env python3
"""Micro-benchmark the non-conv HIP kernels against their ATen equivalents
(BN fwd/bwd, softmax-xent, fused SGD, pooling, accuracy) at the flagship
shapes. Run on an MI355X box; counterpart of tools/bench_conv.py."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def timeit(fn, warmup=10, iters=50):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def row(name, t_ref, t_nat):
    print(f"{name:>34} {t_ref:10.1f} {t_nat:9.1f} {t_ref / t_nat:6.2f}")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=256)
    args = p.parse_args()
    from mi355x_ddp import _C
    from mi355x_ddp.ops import FusedSGD, bn_relu
    from mi355x_ddp.core.metrics import accuracy

    N = args.batch
    dev = "cuda"
    print(f"{'kernel':>34} {'ATen us':>10} {'native':>9} {'x':>6}")

    # --- fused BN+ReLU fwd+bwd, NHWC bf16, stage-1 shape -------------------
    for C, HW in ((64, 32), (256, 8)):
        x = torch.randn(N, C, HW, HW, device=dev, dtype=torch.bfloat16,
                        requires_grad=True).to(memory_format=torch.channels_last)
        bn = torch.nn.BatchNorm2d(C).to(dev)
        dy = torch.randn_like(x)

        def aten():
            y = F.relu(F.batch_norm(x, bn.running_mean, bn.running_var,
                                    bn.weight, bn.bias, True, 0.1, 1e-5))
            torch.autograd.grad(y, [x, bn.weight, bn.bias], dy)

        def native():
            y = bn_relu(x, bn)
            torch.autograd.grad(y, [x, bn.weight, bn.bias], dy)

        row(f"BN+ReLU f/b {N}x{C}x{HW}x{HW}", timeit(aten), timeit(native))

    # --- softmax cross entropy --------------------------------------------
    logits = torch.randn(N, 100, device=dev, dtype=torch.bfloat16,
                         requires_grad=True)
    target = torch.randint(0, 100, (N,), device=dev)
    dy = torch.ones((), device=dev)
    from mi355x_ddp.ops.xent import softmax_cross_entropy
    row("softmax-xent f/b 256x100",
        timeit(lambda: torch.autograd.grad(
            F.cross_entropy(logits, target), [logits], dy)),
        timeit(lambda: torch.autograd.grad(
            softmax_cross_entropy(logits, target), [logits], dy)))

    # --- fused SGD over a ResNet18's parameter set -------------------------
    from mi355x_ddp.models import resnet18
    m1 = resnet18().to(dev)
    m2 = resnet18().to(dev)
    for p_ in list(m1.parameters()) + list(m2.parameters()):
        p_.grad = torch.randn_like(p_)
    ref_opt = torch.optim.SGD(m1.parameters(), lr=1e-9, momentum=0.9,
                              weight_decay=1e-4)
    nat_opt = FusedSGD(m2.parameters(), lr=1e-9, momentum=0.9,
                       weight_decay=1e-4)
    row("SGD step (62 tensors, 11.2M)",
        timeit(ref_opt.step), timeit(nat_opt.step))

    # --- global average pool ----------------------------------------------
    x = torch.randn(N, 512, 4, 4, device=dev, dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    row("global avgpool 256x512x4x4",
        timeit(lambda: F.adaptive_avg_pool2d(x, (1, 1))),
        timeit(lambda: _C.gap_fwd(x)))

    # --- top-k accuracy ----------------------------------------------------
    logits = torch.randn(N, 100, device=dev)

    def aten_acc():
        _, pred = logits.topk(5, 1, True, True)
        pred.t().eq(target.view(1, -1)).float().sum()

    row("top-5 accuracy 256x100",
        timeit(aten_acc), timeit(lambda: accuracy(logits, target)))


if __name__ == "__main__":
    main()
