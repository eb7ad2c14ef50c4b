"""channels_last (NHWC) training path: layout-matched FlatDDP grad views +
layout-agnostic fused SGD must reproduce the NCHW updates exactly."""
import torch

from mi355x_ddp.models import resnet18
from mi355x_ddp.ops import FusedSGD
from mi355x_ddp.parallel import FlatDDP


def _step(model, opt, crit, x, y):
    model.zero_grad_buffer()
    loss = crit(model(x), y)
    loss.backward()
    model.finalize_backward()
    opt.step()
    return float(loss.detach())


def test_channels_last_matches_nchw_training():
    """ReLU-free conv net: NHWC and NCHW training must agree to fp32 roundoff
    (a full ResNet legitimately diverges through borderline ReLU-mask flips
    when the summation order changes with layout)."""
    torch.manual_seed(0)
    data = [(torch.randn(8, 3, 16, 16), torch.randint(0, 10, (8,)))
            for _ in range(3)]

    def make_net():
        torch.manual_seed(1)
        return torch.nn.Sequential(
            torch.nn.Conv2d(3, 16, 3, padding=1),
            torch.nn.Conv2d(16, 32, 3, stride=2, padding=1),
            torch.nn.AdaptiveAvgPool2d((1, 1)),
            torch.nn.Flatten(),
            torch.nn.Linear(32, 10))

    params = {}
    for mode in ("nchw", "nhwc"):
        net = make_net()
        if mode == "nhwc":
            net = net.to(memory_format=torch.channels_last)
        model = FlatDDP(net)
        opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9,
                       weight_decay=1e-4)
        crit = torch.nn.CrossEntropyLoss()
        for x, y in data:
            x = x.to(memory_format=torch.channels_last) if mode == "nhwc" else x
            _step(model, opt, crit, x, y)
        params[mode] = [p.detach().clone().contiguous()
                        for p in model.parameters()]

    for a, b in zip(params["nchw"], params["nhwc"]):
        assert torch.allclose(a, b, atol=1e-5), (a - b).abs().max()


def test_channels_last_resnet_loss_tracks_nchw():
    """Full ResNet18: NHWC loss trajectory tracks NCHW within chaotic drift."""
    torch.manual_seed(0)
    data = [(torch.randn(8, 3, 32, 32), torch.randint(0, 100, (8,)))
            for _ in range(3)]
    losses = {}
    for mode in ("nchw", "nhwc"):
        torch.manual_seed(1)
        net = resnet18()
        if mode == "nhwc":
            net = net.to(memory_format=torch.channels_last)
        model = FlatDDP(net)
        opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9)
        crit = torch.nn.CrossEntropyLoss()
        ls = []
        for x, y in data:
            x = x.to(memory_format=torch.channels_last) if mode == "nhwc" else x
            ls.append(_step(model, opt, crit, x, y))
        losses[mode] = ls
    for i, (a, b) in enumerate(zip(losses["nchw"], losses["nhwc"])):
        assert abs(a - b) < 0.01 + 0.02 * i, losses


def test_channels_last_grad_views_share_layout():
    net = torch.nn.Sequential(torch.nn.Conv2d(3, 8, 3, padding=1),
                              torch.nn.BatchNorm2d(8))
    net = net.to(memory_format=torch.channels_last)
    # static_grads forces the flat-buffer mode at world 1 (the default there
    # is the no-view fast path, which has no flat buffer to test)
    model = FlatDDP(net, static_grads=True)
    conv_w = net[0].weight
    assert conv_w.is_contiguous(memory_format=torch.channels_last)
    assert conv_w.grad.stride() == conv_w.stride()
    # grad view still aliases the flat buffer
    model.flat_grads.fill_(3.0)
    assert torch.all(conv_w.grad == 3.0)
