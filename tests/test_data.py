import torch

from mi355x_ddp.config import TrainConfig
from mi355x_ddp.data import SyntheticCIFAR, build_loaders
from mi355x_ddp.data.cifar import normalize, random_crop_padded


def test_synthetic_shapes_and_determinism():
    ds = SyntheticCIFAR(n=100, seed=7)
    img, label = ds[3]
    assert img.shape == (3, 32, 32)
    assert 0 <= label < 100
    img2, label2 = ds[3]
    assert torch.equal(img, img2) and label == label2
    img3, _ = ds[4]
    assert not torch.equal(img, img3)


def test_eval_set_not_augmented():
    ds = SyntheticCIFAR(n=10, train=False)
    a, _ = ds[0]
    b, _ = ds[0]
    assert torch.equal(a, b)


def test_random_crop_padded_shape():
    img = torch.rand(3, 32, 32)
    out = random_crop_padded(img, 32, 4, gen=torch.Generator().manual_seed(0))
    assert out.shape == (3, 32, 32)


def test_normalize_stats():
    x = torch.rand(3, 32, 32)
    y = normalize(x)
    assert y.shape == x.shape
    assert not torch.equal(x, y)


def test_build_loaders_single_process():
    cfg = TrainConfig(batch_size=32, num_workers=0, synthetic=True)
    train_loader, test_loader, sampler = build_loaders(cfg, 1, 0,
                                                       distributed=False)
    assert sampler is None
    imgs, labels = next(iter(train_loader))
    assert imgs.shape == (32, 3, 32, 32)
    assert labels.shape == (32,)


def test_build_loaders_sharded_counts():
    cfg = TrainConfig(batch_size=32, num_workers=0, synthetic=True)
    l0, _, s0 = build_loaders(cfg, 2, 0, distributed=True)
    l1, _, s1 = build_loaders(cfg, 2, 1, distributed=True)
    assert len(l0) == len(l1)
    assert l0.batch_size == 16  # global 32 / world 2
