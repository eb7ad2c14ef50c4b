import pytest
import torch

from mi355x_ddp.models import build_model, resnet18, resnet34, resnet50


@pytest.mark.parametrize("arch,expansion", [("resnet18", 1), ("resnet34", 1),
                                            ("resnet50", 4)])
def test_forward_shapes(arch, expansion):
    model = build_model(arch, num_classes=100)
    x = torch.randn(4, 3, 32, 32)
    y = model(x)
    assert y.shape == (4, 100)
    assert model.fc.in_features == 512 * expansion


def test_resnet18_param_count():
    # CIFAR ResNet18 w/ 100 classes: ~11.22 M params
    n = sum(p.numel() for p in resnet18().parameters())
    assert 11.0e6 < n < 11.5e6


def test_backward_flows():
    model = resnet18()
    x = torch.randn(2, 3, 32, 32)
    loss = model(x).sum()
    loss.backward()
    grads = [p.grad for p in model.parameters()]
    assert all(g is not None for g in grads)
    assert all(torch.isfinite(g).all() for g in grads)


def test_eval_mode_uses_running_stats():
    model = resnet18()
    model.eval()
    x = torch.randn(2, 3, 32, 32)
    y1 = model(x)
    y2 = model(x)
    assert torch.allclose(y1, y2)


def test_state_dict_roundtrip():
    m1 = resnet34()
    m2 = resnet34()
    m2.load_state_dict(m1.state_dict())
    x = torch.randn(2, 3, 32, 32)
    m1.eval(), m2.eval()
    assert torch.allclose(m1(x), m2(x))


def test_deep_zoo_shapes():
    from mi355x_ddp.models import resnet101, resnet152
    for f, blocks in ((resnet101, 33), (resnet152, 50)):
        m = f(num_classes=100)
        n_bottleneck = sum(type(x).__name__ == "Bottleneck" for x in m.modules())
        assert n_bottleneck == blocks
        y = m(torch.randn(2, 3, 32, 32))
        assert y.shape == (2, 100)


def test_imagenet_stem_variants():
    """Beyond the reference zoo: ImageNet-topology stems (7x7 s2 + maxpool,
    4x spatial reduction before stage 1)."""
    import torch
    from mi355x_ddp.models import build_model
    m = build_model("resnet50_imagenet")
    y = m(torch.randn(2, 3, 224, 224))
    assert y.shape == (2, 100)
    # stage-1 input is 56x56 for 224px inputs (7x7 s2 -> 112, maxpool -> 56)
    feats = {}
    def grab(mod, inp):
        feats["hw"] = inp[0].shape[-2:]
    h = m.layer1.register_forward_pre_hook(grab)
    m(torch.randn(1, 3, 224, 224))
    h.remove()
    assert tuple(feats["hw"]) == (56, 56)
    m18 = build_model("resnet18_imagenet")
    assert m18(torch.randn(1, 3, 96, 96)).shape == (1, 100)
