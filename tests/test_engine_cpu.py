"""Single-process engine tests: training reduces loss, validation and
checkpoint/resume round-trip, grad accumulation equivalence."""
import os

import pytest
import torch

from mi355x_ddp.config import TrainConfig
from mi355x_ddp.core.checkpoint import load_checkpoint, save_checkpoint
from mi355x_ddp.core.engine import train_one_epoch, validate
from mi355x_ddp.core.worker import build_training, init_seeds
from mi355x_ddp.data import build_loaders


def _tiny_cfg(**kw):
    base = dict(batch_size=16, epochs=1, num_workers=0, log_interval=100,
                synthetic=True, lr=0.05)
    base.update(kw)
    return TrainConfig(**base)


def test_loss_decreases_on_fixed_batch():
    init_seeds(0)
    cfg = _tiny_cfg()
    device = torch.device("cpu")
    model, crit, opt, sched, scaler = build_training(
        cfg, device, 1, 0, distributed=True, wrap="flat")
    x = torch.randn(16, 3, 32, 32)
    y = torch.randint(0, 100, (16,))
    model.train()
    losses = []
    for _ in range(12):
        model.zero_grad_buffer()
        loss = crit(model(x), y)
        loss.backward()
        model.finalize_backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] * 0.7, losses


def test_train_one_epoch_and_validate_run():
    init_seeds(0)
    cfg = _tiny_cfg()
    device = torch.device("cpu")
    model, crit, opt, sched, scaler = build_training(
        cfg, device, 1, 0, distributed=True, wrap="flat")
    train_loader, test_loader, _ = build_loaders(cfg, 1, 0, distributed=False)
    avg = train_one_epoch(model, train_loader, crit, opt, 0, cfg, device,
                          max_steps=3)
    assert avg > 0
    acc = validate(model, test_loader, crit, device, cfg)
    assert 0.0 <= acc <= 100.0


def test_checkpoint_roundtrip(tmp_path):
    init_seeds(0)
    cfg = _tiny_cfg(ckpt_dir=str(tmp_path))
    device = torch.device("cpu")
    model, crit, opt, sched, scaler = build_training(
        cfg, device, 1, 0, distributed=True, wrap="flat")
    # one step so optimizer state exists
    x, y = torch.randn(4, 3, 32, 32), torch.randint(0, 100, (4,))
    model.zero_grad_buffer()
    crit(model(x), y).backward()
    model.finalize_backward()
    opt.step()
    path = save_checkpoint(str(tmp_path), cfg.arch, epoch=4, model=model,
                           optimizer=opt, scheduler=sched)
    assert path and os.path.exists(path)

    model2, crit2, opt2, sched2, _ = build_training(
        cfg, device, 1, 0, distributed=True, wrap="flat")
    state = load_checkpoint(path, model2, opt2, sched2)
    assert state["epoch"] == 4
    for p1, p2 in zip(model.parameters(), model2.parameters()):
        assert torch.equal(p1, p2)
    # momentum buffers restored
    s1 = opt.state_dict()["state"]
    s2 = opt2.state_dict()["state"]
    assert len(s1) == len(s2) > 0


def test_grad_accumulation_matches_full_batch():
    """grad_accu_steps=2 over batch B == one step over B for a BN-free model
    (with BatchNorm the batch statistics legitimately differ per micro-batch)."""
    from mi355x_ddp.ops import FusedSGD
    from mi355x_ddp.parallel import FlatDDP
    device = torch.device("cpu")
    torch.manual_seed(1)
    x = torch.randn(8, 3, 8, 8)
    y = torch.randint(0, 10, (8,))

    results = []
    for accu in (1, 2):
        torch.manual_seed(2)
        net = torch.nn.Sequential(
            torch.nn.Conv2d(3, 4, 3), torch.nn.ReLU(), torch.nn.Flatten(),
            torch.nn.Linear(4 * 6 * 6, 10))
        model = FlatDDP(net)
        opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9)
        crit = torch.nn.CrossEntropyLoss()
        model.zero_grad_buffer()
        sub = 8 // accu
        for a in range(accu):
            loss = crit(model(x[a * sub:(a + 1) * sub]),
                        y[a * sub:(a + 1) * sub]) / accu
            loss.backward()
        model.finalize_backward()
        opt.step()
        results.append([p.detach().clone() for p in model.parameters()])
    for p1, p2 in zip(results[0], results[1]):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_fp16_scaler_end_to_end():
    init_seeds(0)
    cfg = _tiny_cfg(amp="fp32")  # CPU: run scaler machinery with fp32 math
    device = torch.device("cpu")
    model, crit, opt, sched, _ = build_training(
        cfg, device, 1, 0, distributed=True, wrap="flat")
    from mi355x_ddp.core.amp import DynamicLossScaler
    scaler = DynamicLossScaler(init_scale=8.0)
    x, y = torch.randn(4, 3, 32, 32), torch.randint(0, 100, (4,))
    model.zero_grad_buffer()
    loss = crit(model(x), y)
    scaler.scale_loss(loss).backward()
    model.finalize_backward()
    grads = [model.flat_grads] if model.flat_grads is not None else \
        [p.grad for p in model.parameters() if p.grad is not None]
    scaler.unscale_(grads)
    assert not scaler.found_inf
    assert scaler.step(opt)


def test_checkpoint_cross_wrap_compat(tmp_path):
    """A checkpoint saved from a FlatDDP-wrapped model loads into a plain
    model and vice versa (state dicts are always the unwrapped module's)."""
    import torch
    from mi355x_ddp.core.checkpoint import load_checkpoint, save_checkpoint
    from mi355x_ddp.models import resnet18
    from mi355x_ddp.parallel import FlatDDP

    torch.manual_seed(0)
    wrapped = FlatDDP(resnet18())
    path = save_checkpoint(str(tmp_path), "resnet18", 3, wrapped)
    assert path is not None

    plain = resnet18()
    state = load_checkpoint(path, plain)
    assert state["epoch"] == 3
    for a, b in zip(plain.state_dict().values(),
                    wrapped.module.state_dict().values()):
        assert torch.equal(a, b)

    # and back: plain-saved -> wrapped load
    path2 = save_checkpoint(str(tmp_path), "resnet18", 4, plain)
    wrapped2 = FlatDDP(resnet18())
    load_checkpoint(path2, wrapped2)
    for a, b in zip(wrapped2.module.state_dict().values(),
                    plain.state_dict().values()):
        assert torch.equal(a, b)


def test_same_seed_reproduces_losses():
    """Determinism: identical seeds give identical loss trajectories on CPU
    (per-rank seeding semantics, reference distributed_mp.py:29-39)."""
    import torch
    from mi355x_ddp.config import TrainConfig
    from mi355x_ddp.core.worker import build_training, init_seeds

    def run():
        init_seeds(7)
        cfg = TrainConfig(batch_size=8, amp="fp32", sync_bn=False)
        device = torch.device("cpu")
        model, crit, opt, sched, _ = build_training(cfg, device, 1, 0,
                                                    distributed=True,
                                                    wrap="flat")
        gen = torch.Generator().manual_seed(5)
        losses = []
        for _ in range(3):
            x = torch.randn(8, 3, 32, 32, generator=gen)
            y = torch.randint(0, 100, (8,), generator=gen)
            model.zero_grad_buffer()
            loss = crit(model(x), y)
            loss.backward()
            model.finalize_backward()
            opt.step()
            losses.append(float(loss))
        return losses

    assert run() == run()


def test_step_watchdog_dumps_on_stall():
    """A step exceeding the threshold dumps all-thread stacks to stderr;
    fast steps dump nothing (SURVEY 5.3 failure diagnostics)."""
    import subprocess
    import sys
    import textwrap
    code = textwrap.dedent("""
        import time
        from mi355x_ddp.core.watchdog import StepWatchdog
        wd = StepWatchdog(0.3)
        with wd:
            time.sleep(1.0)    # stalled 'step'
        with StepWatchdog(5.0):
            time.sleep(0.01)   # fast step
        print("done")
    """)
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=60)
    assert r.returncode == 0 and "done" in r.stdout
    assert "Timeout" in r.stderr and r.stderr.count("Timeout") == 1


def test_watchdog_none_is_noop():
    from mi355x_ddp.core.watchdog import StepWatchdog
    with StepWatchdog(None):
        pass  # must not arm faulthandler at all


def test_best_checkpoint_saved(tmp_path):
    """fit() persists {arch}_best.pt when validation improves."""
    init_seeds(0)
    cfg = _tiny_cfg(ckpt_dir=str(tmp_path), metrics_dir=str(tmp_path),
                    max_train_steps=2, max_eval_steps=2, save_epoch=0)
    device = torch.device("cpu")
    model, crit, opt, sched, scaler = build_training(
        cfg, device, 1, 0, distributed=True, wrap="flat")
    from mi355x_ddp.core.engine import fit
    from mi355x_ddp.data import build_loaders
    tl, vl, _ = build_loaders(cfg, 1, 0, distributed=False)
    fit(model, tl, vl, None, crit, opt, sched, cfg, device, scaler=scaler)
    assert (tmp_path / "resnet18_best.pt").exists()
    assert (tmp_path / "resnet18_final.pt").exists()
