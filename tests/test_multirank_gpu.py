"""Multi-rank training paths ON GPU hardware, within a 1-GPU lease.

RCCL refuses two ranks on one device, so these spawn 2 processes that share
cuda:0 over the gloo backend: the cross-rank reduction math (FlatDDP bucket
all-reduce, SyncBN packed-stat all-reduce, bf16 comm buckets) runs against
tensors produced by the NATIVE HIP kernels — the combination round 1 never
executed on hardware (CPU gloo tests covered the math, single-process GPU
tests covered the kernels; this covers both at once).

SURVEY §4 "multi-GPU integration"; reference wraps at distributed.py:59-60.
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn as nn

pytestmark = pytest.mark.gpu

WORLD = 2


def _run(fn, free_port, world=WORLD):
    mp.spawn(_worker, nprocs=world, args=(world, free_port, fn))


def _worker(rank, world, port, fn):
    torch.cuda.set_device(0)  # both ranks share the single leased GPU
    dist.init_process_group(
        backend="gloo", init_method=f"tcp://127.0.0.1:{port}",
        world_size=world, rank=rank)
    try:
        fn(rank, world)
    finally:
        dist.destroy_process_group()


def _build_native_model(seed: int):
    """Small conv net on cuda:0 with the native conv + fused BN path."""
    from mi355x_ddp.ops import MI355Conv2d
    torch.manual_seed(seed)
    net = nn.Sequential(
        nn.Conv2d(3, 64, 3, padding=1, bias=False),
        nn.BatchNorm2d(64),
        nn.ReLU(),
        nn.Conv2d(64, 64, 3, stride=2, padding=1, bias=False),
        nn.BatchNorm2d(64),
        nn.ReLU(),
        nn.AdaptiveAvgPool2d((1, 1)),
        nn.Flatten(),
        nn.Linear(64, 10),
    ).cuda()
    net = MI355Conv2d.convert(net)
    return net.to(memory_format=torch.channels_last)


def _check_flatddp_native_overlap(rank, world):
    """FlatDDP overlap hooks + native conv/BN kernels + cross-rank averaging:
    2-rank grads equal the single-process full-batch grads."""
    from mi355x_ddp.ops.batchnorm import MI355SyncBatchNorm
    from mi355x_ddp.parallel import FlatDDP
    model = _build_native_model(seed=10)
    # SyncBN so per-rank batch statistics match the full-batch reference
    model = MI355SyncBatchNorm.convert_sync_batchnorm(model)
    wrapped = FlatDDP(model, bucket_cap_mb=1e-4, overlap=True)  # many buckets

    gen = torch.Generator().manual_seed(3)
    xs = [torch.randn(4, 3, 32, 32, generator=gen).cuda()
          .to(memory_format=torch.channels_last) for _ in range(world)]
    ys = [torch.randint(0, 10, (4,), generator=gen).cuda()
          for _ in range(world)]

    wrapped.zero_grad_buffer()
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = wrapped(xs[rank])
        loss = nn.functional.cross_entropy(out, ys[rank])
    loss.backward()
    wrapped.finalize_backward()
    torch.cuda.synchronize()

    # reference: same model, full batch, single process
    ref = _build_native_model(seed=10)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        rout = ref(torch.cat(xs))
        rloss = nn.functional.cross_entropy(rout, torch.cat(ys))
    rloss.backward()
    torch.cuda.synchronize()

    for p, rp in zip(model.parameters(), ref.parameters()):
        a, b = p.grad.float(), rp.grad.float()
        rel = (a - b).norm() / b.norm().clamp_min(1e-8)
        # bf16 forward/backward at batch 4/rank vs batch 8 reference: equal
        # in exact arithmetic, but bf16 rounding differs between the two
        # batch splits; measured drift peaks just over 5% on the smallest
        # tensors. The exact cross-rank math is covered by the fp32 CPU
        # parity tests — this test's job is the native-kernel + collective
        # WIRING on hardware.
        assert rel < 1e-1, f"grad rel err {rel:.4f}"


def _check_syncbn_native_2rank(rank, world):
    """MI355SyncBatchNorm's ONE packed all-reduce per fwd/bwd with native
    stat kernels matches big-batch BatchNorm on GPU."""
    from mi355x_ddp.ops.batchnorm import MI355SyncBatchNorm
    torch.manual_seed(5)
    bn = MI355SyncBatchNorm(64).cuda()
    ref = nn.BatchNorm2d(64).cuda()

    gen = torch.Generator().manual_seed(11)
    xs = [torch.randn(3, 64, 8, 8, generator=gen).cuda()
          .to(memory_format=torch.channels_last) for _ in range(world)]
    x_local = xs[rank].clone().requires_grad_(True)
    x_full = torch.cat(xs).requires_grad_(True)

    y = bn(x_local)
    y_ref = ref(x_full)
    torch.cuda.synchronize()
    assert torch.allclose(y, y_ref[rank * 3:(rank + 1) * 3], atol=1e-4)
    assert torch.allclose(bn.running_mean, ref.running_mean, atol=1e-5)
    assert torch.allclose(bn.running_var, ref.running_var, atol=1e-5)

    y.pow(2).sum().backward()
    y_ref.pow(2).sum().backward()
    torch.cuda.synchronize()
    assert torch.allclose(x_local.grad, x_full.grad[rank * 3:(rank + 1) * 3],
                          atol=1e-3)


def _check_fp16_scaler_2rank_step(rank, world):
    """Full fp16 train step across 2 ranks: autocast fp16, scaled backward,
    native multi-tensor unscale + inf check, fused SGD — ranks end with
    IDENTICAL parameters."""
    from mi355x_ddp.core.amp import DynamicLossScaler
    from mi355x_ddp.ops import FusedSGD
    from mi355x_ddp.parallel import FlatDDP
    model = _build_native_model(seed=20)
    wrapped = FlatDDP(model, overlap=True)
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9,
                   weight_decay=1e-4)
    scaler = DynamicLossScaler(init_scale=1024.0)

    gen = torch.Generator().manual_seed(7)
    for _ in range(2):
        xs = [torch.randn(4, 3, 32, 32, generator=gen).cuda()
              .to(memory_format=torch.channels_last) for _ in range(world)]
        ys = [torch.randint(0, 10, (4,), generator=gen).cuda()
              for _ in range(world)]
        wrapped.zero_grad_buffer()
        with torch.autocast("cuda", dtype=torch.float16):
            loss = nn.functional.cross_entropy(wrapped(xs[rank]), ys[rank])
        scaler.scale_loss(loss).backward()
        wrapped.finalize_backward()
        scaler.unscale_([wrapped.flat_grads])
        assert not scaler.found_inf
        assert scaler.step(opt)
    torch.cuda.synchronize()

    # every rank must hold the same parameters after synced steps
    for p in model.parameters():
        flat = p.detach().float().reshape(-1).cpu()  # gloo all_gather: host
        gathered = [torch.zeros_like(flat) for _ in range(world)]
        dist.all_gather(gathered, flat)
        assert torch.equal(gathered[0], gathered[1])


def _check_bf16_comm_native(rank, world):
    """bf16 comm buckets on GPU tensors: ranks converge to identical grads."""
    from mi355x_ddp.parallel import FlatDDP
    model = _build_native_model(seed=30)
    wrapped = FlatDDP(model, bucket_cap_mb=1e-4, overlap=True,
                      comm_dtype=torch.bfloat16)
    gen = torch.Generator().manual_seed(13)
    xs = [torch.randn(4, 3, 32, 32, generator=gen).cuda()
          .to(memory_format=torch.channels_last) for _ in range(world)]
    wrapped.zero_grad_buffer()
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = wrapped(xs[rank]).pow(2).mean()
    loss.backward()
    wrapped.finalize_backward()
    torch.cuda.synchronize()
    flat = wrapped.flat_grads.cpu()  # gloo all_gather: host staging
    gathered = [torch.zeros_like(flat) for _ in range(world)]
    dist.all_gather(gathered, flat)
    assert torch.equal(gathered[0], gathered[1])


def test_2rank_gpu_flatddp_native_overlap(free_port):
    _run(_check_flatddp_native_overlap, free_port)


def test_2rank_gpu_syncbn_native(free_port):
    _run(_check_syncbn_native_2rank, free_port)


def test_2rank_gpu_fp16_scaler_step(free_port):
    _run(_check_fp16_scaler_2rank_step, free_port)


def test_2rank_gpu_bf16_comm(free_port):
    _run(_check_bf16_comm_native, free_port)
