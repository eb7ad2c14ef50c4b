"""Implicit-GEMM conv kernel numerics vs torch fp32 reference (MI355X).

Covers every conv shape in the CIFAR ResNet18/50 zoo:
3x3 s1, 3x3 s2, 1x1 s1, 1x1 s2 at C 64..512, plus the 3-channel stem
(generic-gather path). Forward, input-grad and weight-grad each compared to
an fp32 eager conv on the same data.
"""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda"

# (N, C, H, W, K, R, stride, pad)
CONV_SHAPES = [
    (8, 3, 32, 32, 64, 3, 1, 1),       # stem (generic path, C=3)
    (8, 64, 32, 32, 64, 3, 1, 1),      # stage1 3x3
    (8, 64, 32, 32, 128, 3, 2, 1),     # stage2 downsampling 3x3 s2
    (8, 64, 32, 32, 128, 1, 2, 0),     # stage2 shortcut 1x1 s2
    (8, 128, 16, 16, 128, 3, 1, 1),    # stage2 3x3
    (8, 128, 16, 16, 256, 3, 2, 1),    # stage3 3x3 s2
    (8, 256, 8, 8, 256, 3, 1, 1),      # stage3 3x3
    (8, 256, 8, 8, 512, 3, 2, 1),      # stage4 3x3 s2
    (8, 512, 4, 4, 512, 3, 1, 1),      # stage4 3x3
    (8, 64, 32, 32, 256, 1, 1, 0),     # resnet50 1x1 expand
    (8, 256, 8, 8, 1024, 1, 2, 0),     # resnet50 1x1 s2 shortcut
    (3, 64, 32, 32, 64, 3, 1, 1),      # ragged batch -> M % 128 != 0
    (5, 3, 30, 30, 64, 3, 1, 1),       # ragged stem (PADC path, M % 64 != 0)
    (8, 3, 224, 224, 64, 3, 1, 1),     # ImageNet-shape stem (PADC, big M)
    (4, 3, 64, 64, 64, 7, 2, 3),       # 7x7 s2 ImageNet stem (PADC, K=196->256)
]


def _mk(shape):
    N, C, H, W, K, R, stride, pad = shape
    g = torch.Generator(device=DEV).manual_seed(hash(shape) % (2**31))
    x = torch.randn(N, C, H, W, generator=g, device=DEV) \
        .to(memory_format=torch.channels_last)
    w = torch.randn(K, C, R, R, generator=g, device=DEV) \
        .to(memory_format=torch.channels_last) / (C * R * R) ** 0.5
    return x, w, stride, pad


def _rel_err(got, ref):
    return (got.float() - ref).norm() / ref.norm().clamp_min(1e-12)


@pytest.mark.parametrize("shape", CONV_SHAPES)
def test_conv_fwd_parity(shape):
    from mi355x_ddp import _C
    x, w, stride, pad = _mk(shape)
    ref = F.conv2d(x, w, None, stride, pad)
    got = _C.conv_fwd_igemm(x.bfloat16(), w.bfloat16(), stride, pad)
    assert got.shape == ref.shape
    assert got.is_contiguous(memory_format=torch.channels_last)
    assert _rel_err(got, ref) < 2e-2, f"rel err {_rel_err(got, ref):.4f}"


@pytest.mark.parametrize("shape", CONV_SHAPES)
def test_conv_dgrad_parity(shape):
    from mi355x_ddp import _C
    x, w, stride, pad = _mk(shape)
    x.requires_grad_(True)
    y = F.conv2d(x, w, None, stride, pad)
    dy = torch.randn_like(y).to(memory_format=torch.channels_last)
    (y * dy).sum().backward()
    ref = x.grad
    wT = w.bfloat16().contiguous(memory_format=torch.channels_last) \
        .flip(2, 3).permute(1, 2, 3, 0).contiguous()
    got = _C.conv_dgrad_igemm(dy.bfloat16(), wT, x.shape[2], x.shape[3],
                              stride, pad)
    assert got.shape == ref.shape
    assert _rel_err(got, ref) < 2e-2, f"rel err {_rel_err(got, ref):.4f}"


@pytest.mark.parametrize("shape", CONV_SHAPES)
def test_conv_wgrad_parity(shape):
    from mi355x_ddp import _C
    x, w, stride, pad = _mk(shape)
    w.requires_grad_(True)
    y = F.conv2d(x, w, None, stride, pad)
    dy = torch.randn_like(y).to(memory_format=torch.channels_last)
    (y * dy).sum().backward()
    ref = w.grad
    K, C, R, _ = w.shape
    # wgrad emits bf16 (K,C,R,S) channels_last directly
    got = _C.conv_wgrad_igemm(dy.bfloat16(), x.bfloat16(), R, R, stride, pad)
    assert got.shape == ref.shape
    assert got.is_contiguous(memory_format=torch.channels_last)
    assert _rel_err(got, ref) < 2e-2, f"rel err {_rel_err(got, ref):.4f}"


@pytest.mark.parametrize("splits", [1, 2, 32, 128])
def test_conv_wgrad_splits_agree(splits):
    """Every split-K factor reduces to the same weight grad (two-stage
    slab reduce, no atomics) — and the stem's padded-channel path too."""
    from mi355x_ddp import _C
    for shape in [(8, 64, 32, 32, 64, 3, 1, 1), (8, 3, 32, 32, 64, 3, 1, 1)]:
        x, w, stride, pad = _mk(shape)
        y = F.conv2d(x, w, None, stride, pad)
        dy = torch.randn_like(y).to(memory_format=torch.channels_last)
        R = w.shape[2]
        ref = _C.conv_wgrad_igemm(dy.bfloat16(), x.bfloat16(), R, R,
                                  stride, pad, 0)
        got = _C.conv_wgrad_igemm(dy.bfloat16(), x.bfloat16(), R, R,
                                  stride, pad, splits)
        assert _rel_err(got, ref.float()) < 5e-3


@pytest.mark.parametrize("wtile", [1, 2, 3, 4, 5, 6, 8, 9])
def test_conv_wgrad_tile_variants_agree(wtile):
    """v2's 128x128 / 256x64 tiles (register 4x4 transpose + ds_write_b64
    staging) compute the same dw as the v1 64x128 tile."""
    from mi355x_ddp import _C
    shapes = [(8, 64, 32, 32, 64, 3, 1, 1),     # M=64 (v2 falls back bounds)
              (8, 64, 32, 32, 256, 1, 1, 0),    # M=256, N=64 (256x64 target)
              (8, 128, 16, 16, 128, 3, 1, 1),   # M=128, N=1152 (128x128)
              (8, 256, 8, 8, 512, 3, 2, 1)]
    for shape in shapes:
        x, w, stride, pad = _mk(shape)
        y = F.conv2d(x, w, None, stride, pad)
        dy = torch.randn_like(y).to(memory_format=torch.channels_last)
        R = w.shape[2]
        ref = _C.conv_wgrad_igemm(dy.bfloat16(), x.bfloat16(), R, R,
                                  stride, pad, 1, 1)  # v1, single split
        got = _C.conv_wgrad_igemm(dy.bfloat16(), x.bfloat16(), R, R,
                                  stride, pad, 0, wtile)
        assert _rel_err(got, ref.float()) < 5e-3, \
            f"{shape} wtile={wtile}: {_rel_err(got, ref.float()):.4f}"


def test_conv_fwd_dgrad_tile228_agree():
    """128x128 tile (BNT=128) fwd/dgrad match the 64-wide-N tile."""
    from mi355x_ddp import _C
    for shape in [(8, 64, 32, 32, 256, 1, 1, 0),
                  (8, 128, 16, 16, 128, 3, 1, 1),
                  (3, 256, 8, 8, 512, 3, 2, 1)]:
        x, w, stride, pad = _mk(shape)
        xb, wb = x.bfloat16(), w.bfloat16()
        a = _C.conv_fwd_igemm(xb, wb, stride, pad, 128)
        b = _C.conv_fwd_igemm(xb, wb, stride, pad, 228)
        assert torch.equal(a, b), shape
        y = F.conv2d(x, w, None, stride, pad)
        dy = torch.randn_like(y).to(memory_format=torch.channels_last).bfloat16()
        wT = wb.contiguous(memory_format=torch.channels_last) \
            .flip(2, 3).permute(1, 2, 3, 0).contiguous()
        da = _C.conv_dgrad_igemm(dy, wT, x.shape[2], x.shape[3], stride, pad, 128)
        db = _C.conv_dgrad_igemm(dy, wT, x.shape[2], x.shape[3], stride, pad, 228)
        assert torch.equal(da, db), shape


def test_stem_fwd_tile_variants_agree():
    """Stem PADC path: both GEMM-M tiles produce the same output."""
    from mi355x_ddp import _C
    x, w, stride, pad = _mk((8, 3, 32, 32, 64, 3, 1, 1))
    a = _C.conv_fwd_igemm(x.bfloat16(), w.bfloat16(), stride, pad, 64)
    b = _C.conv_fwd_igemm(x.bfloat16(), w.bfloat16(), stride, pad, 128)
    assert torch.equal(a, b)


def test_conv_autograd_function_end_to_end():
    """MI355Conv2d: forward+backward through the autograd Function matches
    an fp32 eager conv within bf16 tolerance."""
    from mi355x_ddp.ops import conv2d
    x, w, stride, pad = _mk((4, 64, 32, 32, 128, 3, 2, 1))
    xb = x.bfloat16().requires_grad_(True)
    wb = w.bfloat16().requires_grad_(True)
    y = conv2d(xb, wb, (stride, stride), (pad, pad))
    dy = torch.randn_like(y)
    (y.float() * dy.float()).sum().backward()

    xr = x.clone().requires_grad_(True)
    wr = w.clone().requires_grad_(True)
    yr = F.conv2d(xr, wr, None, stride, pad)
    (yr * dy.float()).sum().backward()

    assert _rel_err(y, yr) < 2e-2
    assert _rel_err(xb.grad, xr.grad) < 3e-2
    assert _rel_err(wb.grad, wr.grad) < 3e-2


def test_resnet_native_conv_step():
    """One fwd+bwd of ResNet18 with MI355Conv2d under bf16 autocast runs and
    produces finite gradients."""
    from mi355x_ddp.models import resnet18
    from mi355x_ddp.ops import MI355Conv2d
    torch.manual_seed(0)
    model = MI355Conv2d.convert(resnet18().to(DEV)) \
        .to(memory_format=torch.channels_last)
    x = torch.randn(16, 3, 32, 32, device=DEV) \
        .to(memory_format=torch.channels_last)
    labels = torch.randint(0, 100, (16,), device=DEV)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = model(x)
        loss = F.cross_entropy(out, labels)
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
    for p in model.parameters():
        assert p.grad is None or torch.isfinite(p.grad).all()
