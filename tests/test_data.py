"""Synthetic + packed datasets."""
import numpy as np
import torch

from gtopkssgd_amd.data import (PackedTensorDataset, SyntheticAN4,
                                SyntheticImages, SyntheticPTB, make_dataset,
                                make_loader, pack_dataset)


def test_synthetic_images_deterministic():
    d = SyntheticImages("cifar10", 100, seed=3)
    x1, y1 = d[7]
    x2, y2 = d[7]
    assert torch.equal(x1, x2) and y1 == y2
    assert x1.shape == (3, 32, 32)


def test_synthetic_ptb_shapes():
    d = SyntheticPTB(vocab_size=50, seq_len=10, n_samples=5)
    toks, tgts = d[0]
    assert toks.shape == (10,) and tgts.shape == (10,)
    assert torch.equal(toks[1:], tgts[:-1])


def test_synthetic_an4():
    d = SyntheticAN4(n_samples=3)
    x, y = d[1]
    assert x.shape[0] == 1 and y.min() >= 1


def test_distributed_sampler_partitions():
    d = make_dataset("mnist", 64)
    l0 = make_loader(d, 8, rank=0, world=2, num_workers=0)
    l1 = make_loader(d, 8, rank=1, world=2, num_workers=0)
    assert len(l0) == len(l1) == 4  # 32 samples per rank / 8


def test_packed_dataset_roundtrip(tmp_path):
    imgs = np.random.rand(10, 3, 8, 8).astype(np.float32)
    labs = np.arange(10, dtype=np.int64)
    prefix = str(tmp_path / "pack")
    pack_dataset(imgs, labs, prefix)
    d = PackedTensorDataset(prefix)
    assert len(d) == 10
    x, y = d[4]
    assert y == 4
    assert torch.allclose(x, torch.from_numpy(imgs[4]))
