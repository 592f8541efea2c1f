"""Synthetic datasets shaped like the reference's real ones.

The sandbox has no network, so MNIST / CIFAR-10 / ImageNet / PTB / AN4
are represented by random tensors of identical shape and label space
(BASELINE.json: "synthetic data of that shape, random-init weights").
Reference parity: `upstream:datasets.py` provided the HDF5 ImageNet
wrapper; the equivalent capability here is deterministic, seeded,
rank-partitioned synthetic data.
"""
from __future__ import annotations

from typing import Tuple

import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler

_SHAPES = {
    "mnist": ((1, 28, 28), 10),
    "cifar10": ((3, 32, 32), 10),
    "imagenet": ((3, 224, 224), 1000),
}


class SyntheticImages(Dataset):
    """Random images + labels with a per-index deterministic generator.

    ``learnable=True`` draws each image around a fixed per-class mean
    direction (SNR ~1), giving a task where loss/accuracy curves are
    meaningful — used by the convergence-comparison harness
    (benchmarks/convergence.py) to reproduce the reference's
    gTop-k-vs-dense validation methodology without real datasets.
    """

    def __init__(self, dataset: str = "cifar10", n_samples: int = 50000,
                 seed: int = 0, learnable: bool = False):
        self.shape, self.num_classes = _SHAPES[dataset]
        self.n = n_samples
        self.seed = seed
        self.learnable = learnable
        if learnable:
            g = torch.Generator().manual_seed(seed * 977 + 13)
            self.class_means = torch.randn(
                (self.num_classes,) + self.shape, generator=g)

    def __len__(self):
        return self.n

    def __getitem__(self, i) -> Tuple[torch.Tensor, int]:
        g = torch.Generator().manual_seed(self.seed * 1_000_003 + i)
        y = int(torch.randint(0, self.num_classes, (1,), generator=g))
        x = torch.randn(self.shape, generator=g)
        if self.learnable:
            x = x + self.class_means[y]
        return x, y


class SyntheticPTB(Dataset):
    """Random token streams shaped like PTB BPTT batches.

    Each item is (tokens[seq_len], targets[seq_len])."""

    def __init__(self, vocab_size: int = 10000, seq_len: int = 35,
                 n_samples: int = 20000, seed: int = 0):
        self.vocab_size = vocab_size
        self.seq_len = seq_len
        self.n = n_samples
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        g = torch.Generator().manual_seed(self.seed * 7_000_003 + i)
        toks = torch.randint(0, self.vocab_size, (self.seq_len + 1,),
                             generator=g)
        return toks[:-1], toks[1:]


class SyntheticAN4(Dataset):
    """Random spectrograms + random transcripts for the CTC path.

    ``learnable=True`` renders each transcript label as a distinctive
    per-label frequency pattern over consecutive time segments (SNR ~2),
    so CTC loss / WER / CER curves are meaningful — the AN4 analogue of
    SyntheticImages' class-mean task."""

    def __init__(self, freq_bins: int = 161, time_steps: int = 200,
                 max_transcript: int = 20, n_labels: int = 29,
                 n_samples: int = 948, seed: int = 0,
                 learnable: bool = False):
        self.freq_bins = freq_bins
        self.time_steps = time_steps
        self.max_transcript = max_transcript
        self.n_labels = n_labels
        self.n = n_samples
        self.seed = seed
        self.learnable = learnable
        if learnable:
            g = torch.Generator().manual_seed(seed * 31 + 7)
            self.label_patterns = torch.randn(n_labels, freq_bins,
                                              generator=g)

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        g = torch.Generator().manual_seed(self.seed * 13_000_003 + i)
        lo = max(1, min(5, self.max_transcript - 1))
        tlen = int(torch.randint(lo, self.max_transcript, (1,), generator=g))
        # labels 1..n_labels-1 (0 is the CTC blank)
        y = torch.randint(1, self.n_labels, (tlen,), generator=g)
        if not self.learnable:
            x = torch.randn(1, self.freq_bins, self.time_steps, generator=g)
            return x, y
        x = torch.randn(1, self.freq_bins, self.time_steps,
                        generator=g) * 0.5
        seg = self.time_steps // tlen
        for j, lab in enumerate(y.tolist()):
            x[0, :, j * seg:(j + 1) * seg] += \
                self.label_patterns[lab].unsqueeze(1)
        return x, y


def make_dataset(dataset: str, n_samples: int = None, seed: int = 0,
                 **kwargs):
    dataset = dataset.lower()
    if dataset in _SHAPES:
        return SyntheticImages(dataset, n_samples or 50000, seed)
    if dataset == "ptb":
        return SyntheticPTB(n_samples=n_samples or 20000, seed=seed,
                            **kwargs)
    if dataset == "an4":
        return SyntheticAN4(n_samples=n_samples or 948, seed=seed, **kwargs)
    if dataset == "an4wav":
        # full audio front-end: tone-sequence waveforms through the real
        # STFT log-spectrogram pipeline (data/audio.py)
        from .audio import SyntheticAN4Waveforms
        return SyntheticAN4Waveforms(n_samples=n_samples or 948, seed=seed,
                                     **kwargs)
    raise ValueError(f"unknown dataset '{dataset}'")


def _ctc_collate(batch):
    """Collate for variable-length CTC transcripts: stack spectrograms,
    keep the label sequences as a list (the trainer flattens them)."""
    xs = torch.stack([b[0] for b in batch])
    ys = [b[1] for b in batch]
    return xs, ys


def make_loader(dataset, batch_size: int, rank: int = 0, world: int = 1,
                num_workers: int = 2, drop_last: bool = True,
                seed: int = 0) -> DataLoader:
    sampler = None
    if world > 1:
        sampler = DistributedSampler(dataset, num_replicas=world, rank=rank,
                                     shuffle=True, seed=seed,
                                     drop_last=drop_last)
    from .audio import SpectrogramDataset, SyntheticAN4Waveforms, \
        ctc_collate_padded
    if isinstance(dataset, (SpectrogramDataset, SyntheticAN4Waveforms)):
        collate = ctc_collate_padded  # variable-length time axis
    elif isinstance(dataset, SyntheticAN4):
        collate = _ctc_collate
    else:
        collate = None
    return DataLoader(dataset, batch_size=batch_size, sampler=sampler,
                      shuffle=(sampler is None), drop_last=drop_last,
                      num_workers=num_workers, pin_memory=True,
                      persistent_workers=num_workers > 0,
                      collate_fn=collate)
