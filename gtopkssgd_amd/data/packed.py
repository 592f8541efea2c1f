"""Packed-tensor dataset for real data without per-sample file IO.

Reference parity: `upstream:datasets.py` `DatasetHDF5` — the reference
packs ImageNet into HDF5 to avoid small-file IO.  h5py is not in this
environment, so the equivalent capability is a pair of memory-mapped
``.npy`` files (images + labels) produced by :func:`pack_dataset`.
Memory-mapping keeps host RSS flat and feeds the DataLoader at NVMe
bandwidth, which is the property the HDF5 wrapper existed for.
"""
from __future__ import annotations

import os
from typing import Tuple

import numpy as np
import torch
from torch.utils.data import Dataset


def pack_dataset(images: np.ndarray, labels: np.ndarray, prefix: str):
    """Write images/labels as <prefix>-images.npy / <prefix>-labels.npy."""
    assert len(images) == len(labels)
    np.save(prefix + "-images.npy", images)
    np.save(prefix + "-labels.npy", labels)


class PackedTensorDataset(Dataset):
    """Memory-mapped packed dataset (images fp32/uint8 NCHW + int labels).

    Optional transform runs per sample on the torch tensor.
    """

    def __init__(self, prefix: str, transform=None):
        img_path = prefix + "-images.npy"
        lab_path = prefix + "-labels.npy"
        if not (os.path.exists(img_path) and os.path.exists(lab_path)):
            raise FileNotFoundError(f"packed dataset not found at {prefix}")
        self.images = np.load(img_path, mmap_mode="r")
        self.labels = np.load(lab_path, mmap_mode="r")
        assert len(self.images) == len(self.labels)
        self.transform = transform

    def __len__(self):
        return len(self.images)

    def __getitem__(self, i) -> Tuple[torch.Tensor, int]:
        x = torch.from_numpy(np.ascontiguousarray(self.images[i]))
        if x.dtype == torch.uint8:
            x = x.float().div_(255.0)
        if self.transform is not None:
            x = self.transform(x)
        return x, int(self.labels[i])
