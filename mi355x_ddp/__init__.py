"""mi355x_ddp — MI355X-native (gfx950/CDNA4) single-node distributed training framework.

Brand-new implementation of the capabilities of rentainhe/pytorch-distributed-training
(reference analysed in SURVEY.md): DDP over RCCL/xGMI, SyncBatchNorm, gradient
accumulation with collective elision, bf16/fp16 mixed precision (apex-equivalent,
no apex), plus hand-written HIP/CDNA4 kernels for the ResNet hot path.

Layout:
    core/      runtime: process group init, engine (train/validate), amp, checkpoint, metrics
    models/    CIFAR ResNet-18/34/50 (reference: utils/model.py) wired to native ops
    ops/       HIP kernel front-ends with CPU fallbacks (csrc/ holds the kernels)
    parallel/  data-parallel wrappers: flat graph-capturable DDP, torch-DDP tuning, DP
    data/      synthetic CIFAR-100-shaped pipeline + real CIFAR-100 when present on disk
"""

__version__ = "0.1.0"

from . import config  # noqa: F401
