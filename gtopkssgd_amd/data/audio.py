"""Audio front-end for the AN4 stack: waveform → log-spectrogram.

Reference parity: SURVEY.md §2.1 "AN4 audio stack" (`audio_data.py` —
``SpectrogramDataset``).  The upstream featurizer follows the
deepspeech.pytorch conventions: 16 kHz audio, 20 ms Hamming window,
10 ms hop, ``log1p(|STFT|)`` features, per-utterance mean/std
normalisation → 161 frequency bins.  This module reproduces that
pipeline with ``torch.stft`` (no librosa in the sandbox) and adds a
deterministic synthetic waveform source so the *whole* chain —
waveform → spectrogram → DeepSpeech → CTC — is exercisable offline.

Featurisation is CPU-side data-loader work (as upstream's was); the
GPU-side hot path starts at the model.
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import torch
from torch.utils.data import Dataset

from ..models.deepspeech import AN4_LABELS

SAMPLE_RATE = 16_000
WINDOW_MS = 20.0
HOP_MS = 10.0


def _n_fft(sample_rate: int = SAMPLE_RATE, window_ms: float = WINDOW_MS):
    return int(sample_rate * window_ms / 1000.0)


def text_to_labels(text: str, labels: str = AN4_LABELS) -> torch.Tensor:
    """Transcript string → int64 label indices (chars not in the label
    set are dropped, as upstream's parser did)."""
    table = {c: i for i, c in enumerate(labels)}
    out = [table[c] for c in text.upper() if c in table]
    return torch.tensor(out, dtype=torch.long)


def labels_to_text(idxs: Sequence[int], labels: str = AN4_LABELS) -> str:
    return "".join(labels[i] for i in idxs)


def spectrogram(waveform: torch.Tensor,
                sample_rate: int = SAMPLE_RATE,
                window_ms: float = WINDOW_MS,
                hop_ms: float = HOP_MS,
                normalize: bool = True) -> torch.Tensor:
    """[num_samples] fp32 waveform → [1, freq_bins, time] log-magnitude
    spectrogram.  freq_bins = n_fft//2 + 1 (161 at the 16 kHz / 20 ms
    defaults, matching ``DeepSpeech(freq_bins=161)``)."""
    n_fft = _n_fft(sample_rate, window_ms)
    hop = int(sample_rate * hop_ms / 1000.0)
    spec = torch.stft(waveform.float(), n_fft=n_fft, hop_length=hop,
                      win_length=n_fft,
                      window=torch.hamming_window(n_fft),
                      center=True, return_complex=True).abs()
    spec = torch.log1p(spec)
    if normalize:
        spec = (spec - spec.mean()) / (spec.std() + 1e-6)
    return spec.unsqueeze(0)


def synth_waveform(transcript: torch.Tensor, seed: int,
                   sample_rate: int = SAMPLE_RATE,
                   char_ms: float = 120.0,
                   snr: float = 2.0) -> torch.Tensor:
    """Render a label sequence as an audible tone sequence: each label
    gets a fixed pair of sinusoid frequencies (a crude two-formant
    'phoneme'), plus white noise.  Deterministic in (transcript, seed).
    This is the waveform-level analogue of SyntheticAN4's learnable
    spectrogram patterns — it makes the synthetic CTC task solvable
    from real STFT features."""
    g = torch.Generator().manual_seed(seed)
    n_char = int(sample_rate * char_ms / 1000.0)
    t = torch.arange(n_char, dtype=torch.float32) / sample_rate
    chunks = []
    for lab in transcript.tolist():
        # two deterministic formants in 200..6000 Hz, unique per label
        f1 = 200.0 + (lab * 211) % 2800
        f2 = 3000.0 + (lab * 157) % 3000
        tone = torch.sin(2 * torch.pi * f1 * t) + \
            0.7 * torch.sin(2 * torch.pi * f2 * t)
        chunks.append(tone)
    wave = torch.cat(chunks) if chunks else torch.zeros(n_char)
    noise = torch.randn(wave.numel(), generator=g) / snr
    return wave + noise


class SpectrogramDataset(Dataset):
    """Upstream-parity dataset: (waveform, transcript string) pairs in,
    (log-spectrogram [1,F,T], label indices) out.

    ``source`` is any sequence of ``(waveform 1-D fp32 tensor, text)``
    pairs — upstream read .wav paths from a manifest csv; the sandbox
    has no audio files, so callers supply the pairs (see
    ``SyntheticAN4Waveforms``)."""

    def __init__(self, source: Sequence[Tuple[torch.Tensor, str]],
                 labels: str = AN4_LABELS,
                 sample_rate: int = SAMPLE_RATE,
                 window_ms: float = WINDOW_MS, hop_ms: float = HOP_MS,
                 normalize: bool = True):
        self.source = source
        self.labels = labels
        self.sample_rate = sample_rate
        self.window_ms = window_ms
        self.hop_ms = hop_ms
        self.normalize = normalize

    def __len__(self):
        return len(self.source)

    def __getitem__(self, i):
        wave, text = self.source[i]
        x = spectrogram(wave, self.sample_rate, self.window_ms,
                        self.hop_ms, self.normalize)
        return x, text_to_labels(text, self.labels)


class SyntheticAN4Waveforms(Dataset):
    """Deterministic waveform-level AN4 stand-in: random transcripts
    rendered to tone-sequence audio, featurised through the real STFT
    pipeline.  Same sizing conventions as ``SyntheticAN4`` (948 train
    utterances upstream)."""

    def __init__(self, n_samples: int = 948, seed: int = 0,
                 max_transcript: int = 20,
                 labels: str = AN4_LABELS):
        self.n = n_samples
        self.seed = seed
        self.max_transcript = max_transcript
        self.labels = labels

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        g = torch.Generator().manual_seed(self.seed * 17_000_003 + i)
        lo = max(1, min(5, self.max_transcript - 1))
        tlen = int(torch.randint(lo, self.max_transcript, (1,),
                                 generator=g))
        y = torch.randint(1, len(self.labels), (tlen,), generator=g)
        wave = synth_waveform(y, seed=self.seed * 23 + i)
        x = spectrogram(wave)
        return x, y


def ctc_collate_padded(batch: List[Tuple[torch.Tensor, torch.Tensor]]):
    """Pad variable-length spectrograms along time to the batch max
    (waveform lengths differ per transcript).  Returns (x [B,1,F,Tmax],
    ys list of label tensors) — same contract as synthetic._ctc_collate
    so DLTrainer's CTC path consumes it unchanged."""
    t_max = max(b[0].shape[-1] for b in batch)
    xs = torch.stack([
        torch.nn.functional.pad(b[0], (0, t_max - b[0].shape[-1]))
        for b in batch])
    ys = [b[1] for b in batch]
    return xs, ys
