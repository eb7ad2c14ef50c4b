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


def test_pickle_cifar100_reader(tmp_path):
    """The torchvision-free CIFAR-100 reader parses the standard python
    pickle layout (fabricated here) and applies the transform chain."""
    import numpy as np
    import pickle
    from mi355x_ddp.data import PickleCIFAR100, build_datasets
    from mi355x_ddp.config import TrainConfig

    d = tmp_path / "cifar-100-python"
    d.mkdir()
    rng = np.random.default_rng(0)
    for name, n in (("train", 12), ("test", 6)):
        payload = {b"data": rng.integers(0, 256, (n, 3072), dtype=np.uint8),
                   b"fine_labels": [int(x) for x in rng.integers(0, 100, n)]}
        with open(d / name, "wb") as f:
            pickle.dump(payload, f)

    ds = PickleCIFAR100(str(tmp_path), train=True)
    assert len(ds) == 12
    img, label = ds[0]
    assert img.shape == (3, 32, 32) and 0 <= label < 100
    img2, label2 = ds[0]
    assert torch.equal(img, img2) and label == label2  # deterministic crop

    # eval set: no augmentation
    ev = PickleCIFAR100(str(tmp_path), train=False)
    assert len(ev) == 6
    a, _ = ev[1]
    b, _ = ev[1]
    assert torch.equal(a, b)

    # build_datasets picks the real data when synthetic=False and it exists
    cfg = TrainConfig(synthetic=False, data_root=str(tmp_path))
    tr, te = build_datasets(cfg)
    assert isinstance(tr, PickleCIFAR100) and len(te) == 6
