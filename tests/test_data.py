"""Data pipeline: synthetic determinism, CIFAR pickle reader, loader
entry points, distributed sampler sharding."""

import os
import pickle

import numpy as np
import torch

from bdbnn_amd.data import (
    SyntheticImageDataset, CIFAR10Dataset, dataloader_cifar10,
    dataloader_imagenet, dataloader_synthetic)


def test_synthetic_deterministic_per_index():
    ds = SyntheticImageDataset(100, (3, 32, 32), 10, seed=7)
    x1, y1 = ds[42]
    x2, y2 = ds[42]
    assert torch.equal(x1, x2) and y1 == y2
    x3, _ = ds[43]
    assert not torch.equal(x1, x3)
    assert x1.shape == (3, 32, 32) and 0 <= y1 < 10


def _write_fake_cifar(root):
    d = os.path.join(root, "cifar-10-batches-py")
    os.makedirs(d)
    rng = np.random.RandomState(0)
    for i in range(1, 6):
        batch = {"data": rng.randint(0, 255, (20, 3072), dtype=np.uint8)
                 .astype(np.uint8),
                 "labels": rng.randint(0, 10, 20).tolist()}
        with open(os.path.join(d, f"data_batch_{i}"), "wb") as f:
            pickle.dump(batch, f)
    batch = {"data": rng.randint(0, 255, (10, 3072), dtype=np.uint8),
             "labels": rng.randint(0, 10, 10).tolist()}
    with open(os.path.join(d, "test_batch"), "wb") as f:
        pickle.dump(batch, f)


def test_cifar_pickle_reader(tmp_path):
    _write_fake_cifar(str(tmp_path))
    ds = CIFAR10Dataset(str(tmp_path), train=True)
    assert len(ds) == 100
    img, label = ds[0]
    assert img.shape == (3, 32, 32) and 0 <= label < 10
    val = CIFAR10Dataset(str(tmp_path), train=False)
    assert len(val) == 10


def test_dataloader_cifar10_real_files(tmp_path):
    _write_fake_cifar(str(tmp_path))
    loader = dataloader_cifar10("train", batch_size=8,
                                data_path=str(tmp_path), workers=0)
    x, y = next(iter(loader))
    assert x.shape == (8, 3, 32, 32)


def test_cifar_tarball_bootstrap(tmp_path):
    """A cifar-10-python.tar.gz in the dataset root is auto-extracted
    (the no-network stand-in for the reference's download=True,
    ref:loader.py:23-26)."""
    import shutil
    import tarfile
    staging = tmp_path / "staging"
    _write_fake_cifar(str(staging))
    root = tmp_path / "data"
    os.makedirs(root)
    with tarfile.open(root / "cifar-10-python.tar.gz", "w:gz") as tf:
        tf.add(staging / "cifar-10-batches-py",
               arcname="cifar-10-batches-py")
    shutil.rmtree(staging)
    loader = dataloader_cifar10("train", batch_size=8,
                                data_path=str(root), workers=0)
    x, y = next(iter(loader))
    assert x.shape == (8, 3, 32, 32)
    assert (root / "cifar-10-batches-py" / "data_batch_1").exists()


def test_dataloader_missing_explicit_path_raises(tmp_path):
    # a wrong --data path must not silently 'train' on noise
    import pytest
    with pytest.raises(FileNotFoundError):
        dataloader_imagenet("train", batch_size=4,
                            data_path=str(tmp_path / "nope"), workers=0)


def test_dataloader_synthetic_when_requested(tmp_path):
    loader = dataloader_imagenet("train", batch_size=4,
                                 data_path=str(tmp_path), workers=0,
                                 synthetic=True, synthetic_len=16)
    x, y = next(iter(loader))
    assert x.shape == (4, 3, 224, 224)


def test_dataloader_no_path_falls_back_with_warning(caplog):
    import logging
    with caplog.at_level(logging.WARNING, logger="bdbnn"):
        loader = dataloader_imagenet("train", batch_size=4, data_path=None,
                                     workers=0, synthetic_len=16)
    assert any("SYNTHETIC" in r.message for r in caplog.records)
    x, y = next(iter(loader))
    assert x.shape == (4, 3, 224, 224)


def test_random_resized_crop_box_semantics():
    # scale in [0.08, 1.0] x area, aspect in [3/4, 4/3] (ref:loader.py:59-64)
    from bdbnn_amd.data.loaders import _random_resized_crop_box
    torch.manual_seed(0)
    areas, aspects = [], []
    for _ in range(300):
        left, top, w, h = _random_resized_crop_box(500, 400)
        assert 0 <= left <= 500 - w and 0 <= top <= 400 - h
        areas.append(w * h / (500 * 400))
        aspects.append(w / h)
    assert min(areas) < 0.25 and max(areas) > 0.5  # spans the scale range
    assert all(0.70 <= a <= 1.40 for a in aspects)  # rounding slack
    assert min(aspects) < 0.85 and max(aspects) > 1.15


def test_synthetic_loader_drop_last():
    loader = dataloader_synthetic(8, (3, 32, 32), 10, length=20, workers=0)
    batches = list(loader)
    assert len(batches) == 2  # drop_last
