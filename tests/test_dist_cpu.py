"""Multi-process distributed plumbing over gloo on CPU (world_size 2).

Covers the runtime paths that must be correct by construction before any GPU
run: reduce_mean, FlatDDP gradient averaging vs a single-process large batch,
no_sync collective elision, SyncBN statistics parity, sampler sharding.
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn as nn

from mi355x_ddp.core.dist import reduce_mean

WORLD = 2


def _run(fn, free_port, world=WORLD):
    mp.spawn(_worker, nprocs=world, args=(world, free_port, fn))


def _worker(rank, world, port, fn):
    dist.init_process_group(
        backend="gloo", init_method=f"tcp://127.0.0.1:{port}",
        world_size=world, rank=rank)
    try:
        fn(rank, world)
    finally:
        dist.destroy_process_group()


# --- payload fns (module-level for pickling) -------------------------------

def _check_reduce_mean(rank, world):
    t = torch.tensor([float(rank + 1)])
    out = reduce_mean(t, world)
    assert torch.allclose(out, torch.tensor([1.5])), out
    assert torch.allclose(t, torch.tensor([float(rank + 1)]))  # input untouched


def _check_flat_ddp_grads(rank, world):
    from mi355x_ddp.parallel import FlatDDP
    torch.manual_seed(42)  # same init on both ranks
    model = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))
    wrapped = FlatDDP(model, bucket_cap_mb=1e-5)  # force multiple buckets

    # reference: single-process over the concatenated batch
    torch.manual_seed(42)
    ref = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))

    gen = torch.Generator().manual_seed(7)
    xs = [torch.randn(4, 8, generator=gen) for _ in range(world)]

    wrapped.zero_grad_buffer()
    out = wrapped(xs[rank])
    out.pow(2).mean().backward()
    wrapped.finalize_backward()

    ref_out = ref(torch.cat(xs))
    ref_out.pow(2).mean().backward()

    for p, rp in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.grad, rp.grad, atol=1e-6), \
            (p.grad - rp.grad).abs().max()


def _check_no_sync_elision(rank, world):
    from mi355x_ddp.parallel import FlatDDP
    torch.manual_seed(0)
    model = nn.Linear(4, 4)
    wrapped = FlatDDP(model)
    x = torch.full((2, 4), float(rank + 1))

    wrapped.zero_grad_buffer()
    with wrapped.no_sync():
        wrapped(x).sum().backward()
        wrapped.finalize_backward()
    # grads must be LOCAL (different across ranks)
    gathered = [torch.zeros_like(wrapped.flat_grads) for _ in range(world)]
    dist.all_gather(gathered, wrapped.flat_grads)
    assert not torch.allclose(gathered[0], gathered[1])

    # second micro-step with sync: grads now averaged and equal
    wrapped(x).sum().backward()
    wrapped.finalize_backward()
    gathered2 = [torch.zeros_like(wrapped.flat_grads) for _ in range(world)]
    dist.all_gather(gathered2, wrapped.flat_grads)
    assert torch.allclose(gathered2[0], gathered2[1], atol=1e-6)


def _check_syncbn(rank, world):
    from mi355x_ddp.ops.batchnorm import MI355SyncBatchNorm
    torch.manual_seed(5)
    bn = MI355SyncBatchNorm(6)
    ref = nn.BatchNorm2d(6)

    gen = torch.Generator().manual_seed(11)
    xs = [torch.randn(3, 6, 4, 4, generator=gen) for _ in range(world)]
    x_local = xs[rank].clone().requires_grad_(True)
    x_full = torch.cat(xs).requires_grad_(True)

    y = bn(x_local)
    y_ref = ref(x_full)
    assert torch.allclose(y, y_ref[rank * 3:(rank + 1) * 3], atol=1e-5)
    assert torch.allclose(bn.running_mean, ref.running_mean, atol=1e-6)
    assert torch.allclose(bn.running_var, ref.running_var, atol=1e-6)

    # backward parity: d/dx of sum(y^2) matches the big-batch reference slice
    y.pow(2).sum().backward()
    y_ref.pow(2).sum().backward()
    assert torch.allclose(x_local.grad, x_full.grad[rank * 3:(rank + 1) * 3],
                          atol=1e-4)
    # dgamma/dbeta are local sums; summed across ranks they equal the reference's
    dist.all_reduce(bn.weight.grad)
    dist.all_reduce(bn.bias.grad)
    assert torch.allclose(bn.weight.grad, ref.weight.grad, atol=1e-4)
    assert torch.allclose(bn.bias.grad, ref.bias.grad, atol=1e-4)


def _check_convert_syncbn(rank, world):
    from mi355x_ddp.models import resnet18
    from mi355x_ddp.ops.batchnorm import MI355SyncBatchNorm
    m = resnet18()
    m = MI355SyncBatchNorm.convert_sync_batchnorm(m)
    n_sync = sum(isinstance(x, MI355SyncBatchNorm) for x in m.modules())
    n_plain = sum(type(x) is nn.BatchNorm2d for x in m.modules())
    assert n_plain == 0 and n_sync == 20  # ResNet18: 20 BN layers
    # forward still works and stays rank-consistent
    y = m(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 100)


# --- tests -----------------------------------------------------------------

def test_reduce_mean(free_port):
    _run(_check_reduce_mean, free_port)


def test_flat_ddp_grad_parity(free_port):
    _run(_check_flat_ddp_grads, free_port)


def test_no_sync_elision(free_port):
    _run(_check_no_sync_elision, free_port)


def test_syncbn_parity(free_port):
    _run(_check_syncbn, free_port)


def test_convert_syncbn(free_port):
    _run(_check_convert_syncbn, free_port)


def test_sampler_shards_are_disjoint():
    from torch.utils.data.distributed import DistributedSampler
    from mi355x_ddp.data import SyntheticCIFAR
    ds = SyntheticCIFAR(n=100)
    s0 = DistributedSampler(ds, num_replicas=2, rank=0, shuffle=True, seed=1)
    s1 = DistributedSampler(ds, num_replicas=2, rank=1, shuffle=True, seed=1)
    s0.set_epoch(3), s1.set_epoch(3)
    i0, i1 = set(iter(s0)), set(iter(s1))
    assert len(i0) == len(i1) == 50
    assert i0.isdisjoint(i1)


def _check_torch_ddp_parity(rank, world):
    """torch's C++ reducer (A/B reference mode) produces the same averaged
    grads as FlatDDP on the same sharded batch."""
    from mi355x_ddp.parallel import FlatDDP, wrap_torch_ddp
    torch.manual_seed(21)
    net_a = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))
    torch.manual_seed(21)
    net_b = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))
    flat = FlatDDP(net_a)
    tddp = wrap_torch_ddp(net_b, device_id=None)

    gen = torch.Generator().manual_seed(9)
    xs = [torch.randn(4, 8, generator=gen) for _ in range(world)]
    flat.zero_grad_buffer()
    flat(xs[rank]).pow(2).mean().backward()
    flat.finalize_backward()
    tddp(xs[rank]).pow(2).mean().backward()
    for p, q in zip(net_a.parameters(), net_b.parameters()):
        assert torch.allclose(p.grad, q.grad, atol=1e-6), \
            (p.grad - q.grad).abs().max()


def _check_grad_accu_matches_large_batch(rank, world):
    """2 ranks x 2 micro-steps with no_sync == one 4-shard averaged batch."""
    from mi355x_ddp.parallel import FlatDDP
    torch.manual_seed(33)
    model = nn.Linear(6, 3)
    wrapped = FlatDDP(model)
    gen = torch.Generator().manual_seed(17)
    micro = [torch.randn(2, 6, generator=gen) for _ in range(2 * world)]

    wrapped.zero_grad_buffer()
    with wrapped.no_sync():
        (wrapped(micro[rank * 2]).pow(2).mean() / 2).backward()
        wrapped.finalize_backward()
    (wrapped(micro[rank * 2 + 1]).pow(2).mean() / 2).backward()
    wrapped.finalize_backward()

    torch.manual_seed(33)
    ref = nn.Linear(6, 3)
    sum(ref(m).pow(2).mean() for m in micro).div(2 * world).backward()
    for p, rp in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p.grad, rp.grad, atol=1e-6)


def test_torch_ddp_parity(free_port):
    _run(_check_torch_ddp_parity, free_port)


def test_grad_accu_multirank_matches_large_batch(free_port):
    _run(_check_grad_accu_matches_large_batch, free_port)


def test_flat_ddp_bucket_invariants():
    """Buckets tile the flat buffer exactly: contiguous, disjoint, complete,
    and every param's grad view lies inside exactly one bucket."""
    from mi355x_ddp.models import resnet18
    from mi355x_ddp.parallel import FlatDDP
    model = FlatDDP(resnet18(), bucket_cap_mb=5, static_grads=True)
    numel = sum(p.numel() for p in model._params)
    assert model.flat_grads.numel() == numel
    prev_end = 0
    for (s, e, ps) in model._buckets:
        assert s == prev_end and e > s
        prev_end = e
    assert prev_end == numel
    covered = set()
    for p in model._params:
        off, view = model._views[p]
        assert view.data_ptr() == model.flat_grads[off:off + p.numel()].data_ptr()
        assert p.grad is view and view.shape == p.shape
        bi = model._param_bucket[p]
        s, e, ps = model._buckets[bi]
        assert s <= off and off + p.numel() <= e and p in ps
        covered.add((off, off + p.numel()))
    spans = sorted(covered)
    for (s1, e1), (s2, e2) in zip(spans, spans[1:]):
        assert e1 <= s2  # disjoint views
    assert spans[0][0] == 0 and spans[-1][1] == numel


def _check_collective_counts(rank, world):
    """no_sync elides EVERY gradient all_reduce; a synced micro-step fires
    one per bucket; SyncBN fires exactly one all_reduce per fwd and per bwd."""
    from mi355x_ddp.core.dist import count_collectives
    from mi355x_ddp.ops.batchnorm import MI355SyncBatchNorm
    from mi355x_ddp.parallel import FlatDDP
    torch.manual_seed(1)
    model = nn.Sequential(nn.Linear(4, 8), nn.Linear(8, 2))
    wrapped = FlatDDP(model, bucket_cap_mb=1e-5)  # several buckets
    nbuckets = len(wrapped._buckets)
    x = torch.randn(2, 4)

    wrapped.zero_grad_buffer()
    with count_collectives() as counts:
        with wrapped.no_sync():
            wrapped(x).sum().backward()
            wrapped.finalize_backward()
    assert counts.get("all_reduce", 0) == 0, counts

    with count_collectives() as counts:
        wrapped(x).sum().backward()
        wrapped.finalize_backward()
    assert counts.get("all_reduce", 0) == nbuckets, (counts, nbuckets)

    bn = MI355SyncBatchNorm(4)
    xb = torch.randn(2, 4, 3, 3, requires_grad=True)
    with count_collectives() as counts:
        y = bn(xb)
    assert counts.get("all_reduce", 0) == 1, counts
    with count_collectives() as counts:
        y.sum().backward()
    assert counts.get("all_reduce", 0) == 1, counts


def test_collective_counts(free_port):
    _run(_check_collective_counts, free_port)


def _check_bf16_comm_parity(rank, world):
    """comm_dtype=bf16 all-reduces half the bytes; grads match the fp32-comm
    path within bf16 rounding."""
    from mi355x_ddp.parallel import FlatDDP
    torch.manual_seed(42)
    model = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))
    wrapped = FlatDDP(model, bucket_cap_mb=1e-5,
                      comm_dtype=torch.bfloat16)
    assert wrapped.comm_grads is not None
    assert wrapped.comm_grads.dtype == torch.bfloat16

    torch.manual_seed(42)
    ref_model = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))
    ref = FlatDDP(ref_model, bucket_cap_mb=1e-5)  # fp32 comm

    gen = torch.Generator().manual_seed(7)
    xs = [torch.randn(4, 8, generator=gen) for _ in range(world)]

    for w in (wrapped, ref):
        w.zero_grad_buffer()
        w(xs[rank]).pow(2).mean().backward()
        w.finalize_backward()

    for p, rp in zip(model.parameters(), ref_model.parameters()):
        # bf16 has ~3 decimal digits; grads here are O(0.1)
        assert torch.allclose(p.grad, rp.grad, atol=2e-3, rtol=2e-2), \
            (p.grad - rp.grad).abs().max()

    # reduce_flat path too
    wrapped.zero_grad_buffer()
    wrapped(xs[rank]).pow(2).mean().backward()
    wrapped.reduce_flat()
    gathered = [torch.zeros_like(wrapped.flat_grads) for _ in range(world)]
    dist.all_gather(gathered, wrapped.flat_grads)
    assert torch.allclose(gathered[0], gathered[1])  # ranks agree exactly


def test_bf16_comm_parity(free_port):
    _run(_check_bf16_comm_parity, free_port)


class _SkipsParam(nn.Module):
    """Static-graph violation: `unused` never receives a gradient."""

    def __init__(self):
        super().__init__()
        self.used = nn.Linear(4, 4)
        self.unused = nn.Linear(4, 4)

    def forward(self, x):
        return self.used(x)


def _check_unused_param_fails_fast(rank, world):
    from mi355x_ddp.parallel import FlatDDP
    torch.manual_seed(0)
    wrapped = FlatDDP(_SkipsParam(), bucket_cap_mb=1e-5)
    wrapped.zero_grad_buffer()
    wrapped(torch.randn(2, 4)).sum().backward()
    try:
        wrapped.finalize_backward()
    except RuntimeError as e:
        assert "static graph" in str(e) and "unused" in str(e), e
    else:
        raise AssertionError("finalize_backward should have raised on the "
                             "never-filled bucket instead of hanging")


def test_unused_param_fails_fast(free_port):
    """FlatDDP raises (listing the parameters) instead of deadlocking when a
    parameter never receives a gradient — the static-graph contract."""
    _run(_check_unused_param_fails_fast, free_port)


def test_world1_fast_path_matches_static():
    """At world 1 FlatDDP installs no grad views (AccumulateGrad assigns,
    zero add kernels); one optimizer step must match the flat-buffer mode."""
    from mi355x_ddp.ops import FusedSGD
    from mi355x_ddp.parallel import FlatDDP

    results = {}
    for mode in ("fast", "static"):
        torch.manual_seed(3)
        net = nn.Sequential(nn.Linear(6, 12), nn.ReLU(), nn.Linear(12, 3))
        model = FlatDDP(net, static_grads=(mode == "static"))
        if mode == "fast":
            assert model.flat_grads is None
            assert all(p.grad is None for p in net.parameters())
        else:
            assert model.flat_grads is not None
        opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9,
                       weight_decay=1e-4)
        gen = torch.Generator().manual_seed(9)
        for _ in range(3):
            x = torch.randn(5, 6, generator=gen)
            model.zero_grad_buffer()
            model(x).pow(2).mean().backward()
            model.finalize_backward()
            opt.step()
        results[mode] = [p.detach().clone() for p in net.parameters()]
    for a, b in zip(results["fast"], results["static"]):
        assert torch.allclose(a, b, atol=1e-7), (a - b).abs().max()


def _check_grad_accu_with_bf16_comm(rank, world):
    """no_sync micro-steps + bf16 comm buckets: accumulation stays fp32 and
    the final synced step matches the fp32-comm path within bf16 wire
    rounding."""
    from mi355x_ddp.parallel import FlatDDP
    results = {}
    for comm in ("bf16", "fp32"):
        torch.manual_seed(21)
        model = nn.Sequential(nn.Linear(6, 12), nn.ReLU(), nn.Linear(12, 3))
        wrapped = FlatDDP(model, bucket_cap_mb=1e-5,
                          comm_dtype=torch.bfloat16 if comm == "bf16" else None)
        gen = torch.Generator().manual_seed(4)
        xs = [torch.randn(4, 6, generator=gen) for _ in range(2 * world)]
        wrapped.zero_grad_buffer()
        with wrapped.no_sync():
            wrapped(xs[rank]).pow(2).mean().backward()
            wrapped.finalize_backward()
        wrapped(xs[world + rank]).pow(2).mean().backward()
        wrapped.finalize_backward()
        results[comm] = wrapped.flat_grads.clone()
    assert torch.allclose(results["bf16"], results["fp32"],
                          atol=5e-3, rtol=5e-2), \
        (results["bf16"] - results["fp32"]).abs().max()


def test_grad_accu_with_bf16_comm(free_port):
    _run(_check_grad_accu_with_bf16_comm, free_port)


def _check_syncbn_with_o2(rank, world):
    """SyncBN with bf16 (O2) parameters: packed-stat all_reduce and the
    backward still match the gathered-batch fp32 reference within bf16
    tolerance, and ranks agree bitwise after the grad all-reduce."""
    from mi355x_ddp.core.amp import cast_model_bf16
    from mi355x_ddp.ops.batchnorm import MI355SyncBatchNorm
    from mi355x_ddp.parallel import FlatDDP
    torch.manual_seed(31)
    net = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1, bias=False),
                        nn.BatchNorm2d(8))
    net = MI355SyncBatchNorm.convert_sync_batchnorm(net)
    net = cast_model_bf16(net)
    wrapped = FlatDDP(net, grad_dtype=torch.bfloat16)
    assert wrapped.flat_grads.dtype == torch.bfloat16
    gen = torch.Generator().manual_seed(17)
    xs = [torch.randn(2, 3, 8, 8, generator=gen).bfloat16()
          for _ in range(world)]
    wrapped.zero_grad_buffer()
    out = wrapped(xs[rank])
    out.float().pow(2).mean().backward()
    wrapped.finalize_backward()
    # ranks agree bitwise after the (bf16) gradient all-reduce
    flat = wrapped.flat_grads.cpu()
    gathered = [torch.zeros_like(flat) for _ in range(world)]
    dist.all_gather(gathered, flat)
    assert torch.equal(gathered[0], gathered[1])
    # running stats were updated and stay finite with bf16 parameters;
    # both ranks hold the SAME stats (they came from the packed all_reduce)
    bn = net[1]
    stats = torch.cat([bn.running_mean.float(), bn.running_var.float()]).cpu()
    assert torch.isfinite(stats).all()
    gathered_s = [torch.zeros_like(stats) for _ in range(world)]
    dist.all_gather(gathered_s, stats)
    assert torch.allclose(gathered_s[0], gathered_s[1], atol=1e-6)
    assert not torch.equal(bn.running_mean.float(),
                           torch.zeros_like(bn.running_mean.float()))


def test_syncbn_with_o2(free_port):
    _run(_check_syncbn_with_o2, free_port)
