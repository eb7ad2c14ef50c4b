"""hipGraph-captured training step (opt-in, `--hip-graph`).

The whole step — zero-grad, forward, loss, backward, flat gradient
all-reduce, fused SGD — is captured once into a hipGraph and replayed per
batch; per-step host work collapses to two H2D copies + one graph launch.
Requires FlatDDP with static_grads=True (param.grad are views of one flat
buffer) and the arithmetic-count BN path (no host syncs inside the step).
RCCL collectives are capturable, so the world_size>1 all-reduce is inside
the graph too.

Measured honestly (round 1, 1x MI355X, ResNet18/CIFAR bf16): graph replay is
~3-5% SLOWER than the eager step at world 1 — the eager launch stream already
overlaps with GPU work at this kernel size, while replay serialises the H2D
static-buffer copies with the replay and pays the ~10-16 us replay floor
(profiles/bench_results_r1.md: 35.2-36.9k img/s graphed vs 38.7-47.2k eager).
It exists because it is the launch-overhead-proof variant: the win shows up
when per-step host work grows (small models at very small batch, or a
host-jittery rank fleet), not on this benchmark. Default stays eager.
"""
from __future__ import annotations

from typing import Optional

import torch

from ..config import TrainConfig
from .amp import autocast_ctx


class GraphedTrainStep:
    def __init__(self, model, criterion, optimizer, cfg: TrainConfig,
                 device: torch.device, batch: int, image_size: int = 32,
                 num_classes: int = 100, warmup_iters: int = 3):
        assert getattr(model, "flat_grads", None) is not None, \
            "GraphedTrainStep needs FlatDDP with static_grads=True " \
            "(static grad memory is what makes the capture replayable)"
        self.model = model
        self.criterion = criterion
        self.optimizer = optimizer
        self.cfg = cfg
        self.device = device
        self.static_img = torch.zeros(batch, 3, image_size, image_size,
                                      device=device)
        if cfg.channels_last:
            self.static_img = self.static_img.to(memory_format=torch.channels_last)
        self.static_lbl = torch.zeros(batch, dtype=torch.long, device=device)
        self.static_loss: Optional[torch.Tensor] = None

        # warmup on a side stream (allocator + autotune settle, momentum
        # buffers materialise), then capture
        self.static_img.normal_()
        self.static_lbl.random_(0, num_classes)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                self._step()
        torch.cuda.current_stream().wait_stream(side)

        self._captured_hp = self._hyperparams()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self._step()

    def _hyperparams(self):
        """Every optimizer hyperparameter baked into captured kernel args.

        Snapshot ALL param groups' scalar settings (lr, momentum, weight
        decay, ...), not just lr — any of them changing between replays means
        the captured step is stale and must be re-captured.
        """
        return tuple(
            tuple(sorted((k, v) for k, v in g.items()
                         if isinstance(v, (int, float, bool))))
            for g in self.optimizer.param_groups)

    def _step(self):
        self.model.zero_grad_buffer()
        with autocast_ctx(self.cfg.amp, "cuda"):
            out = self.model(self.static_img)
            loss = self.criterion(out, self.static_lbl)
        loss.backward()
        self.model.reduce_flat()
        self.optimizer.step()
        self.static_loss = loss.detach()

    def run(self, images: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
        """Copy the batch into the static buffers and replay the graph.

        Kernel arguments (the fused SGD's lr/momentum/weight_decay, any
        added param group's settings) are frozen at capture time, so a change
        in ANY optimizer hyperparameter triggers a re-capture — MultiStepLR
        changes 3 times in 200 epochs, so this is rare and cheap.

        world_size>1 contract: the LR schedule (and any other hyperparameter
        change) must be rank-synchronous — the re-capture replays an RCCL
        collective, so every rank must re-capture at the same step or the
        ranks' collectives desynchronise.
        """
        hp = self._hyperparams()
        if hp != self._captured_hp:
            self._captured_hp = hp
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                self._step()
        self.static_img.copy_(images, non_blocking=True)
        self.static_lbl.copy_(labels, non_blocking=True)
        self.graph.replay()
        return self.static_loss
