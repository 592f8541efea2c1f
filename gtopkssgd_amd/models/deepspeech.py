"""DeepSpeech-style LSTM acoustic model for AN4-shaped audio.

Reference parity: SURVEY.md §2.1 "AN4 audio stack" — a spectrogram-in,
CTC-logits-out bidirectional-LSTM acoustic model plus a greedy CTC
decoder.  Input: [batch, 1, freq_bins, time]; output: [time', batch,
n_labels] suitable for ``nn.CTCLoss``.
"""
from typing import List

import torch
import torch.nn as nn
import torch.nn.functional as F

# AN4 label set: blank + apostrophe + A-Z + space
AN4_LABELS = "_'ABCDEFGHIJKLMNOPQRSTUVWXYZ "


class SequenceBatchNorm(nn.Module):
    """BatchNorm over the feature dim of a [T, B, H] sequence."""

    def __init__(self, hidden: int):
        super().__init__()
        self.bn = nn.BatchNorm1d(hidden)

    def forward(self, x):
        t, b, h = x.shape
        return self.bn(x.reshape(t * b, h)).reshape(t, b, h)


class DeepSpeech(nn.Module):
    def __init__(self, freq_bins: int = 161, hidden: int = 512,
                 num_layers: int = 4, n_labels: int = len(AN4_LABELS)):
        super().__init__()
        self.conv = nn.Sequential(
            nn.Conv2d(1, 32, (41, 11), stride=(2, 2), padding=(20, 5)),
            nn.BatchNorm2d(32), nn.Hardtanh(0, 20, inplace=True),
            nn.Conv2d(32, 32, (21, 11), stride=(2, 1), padding=(10, 5)),
            nn.BatchNorm2d(32), nn.Hardtanh(0, 20, inplace=True))
        conv_out = 32 * (((freq_bins + 1) // 2 + 1) // 2)
        self.rnns = nn.ModuleList()
        self.bns = nn.ModuleList()
        in_size = conv_out
        for _ in range(num_layers):
            self.rnns.append(nn.LSTM(in_size, hidden, bidirectional=True))
            self.bns.append(SequenceBatchNorm(hidden))
            in_size = hidden
        self.fc = nn.Linear(hidden, n_labels)

    def forward(self, x):
        # x: [B, 1, F, T]
        if x.device.type == "cpu" and torch.backends.mkldnn.enabled:
            # oneDNN's conv backward hangs (multi-thread) or segfaults
            # (single-thread) for this (41,11)/pad(20,5) geometry on this
            # torch build.  convolution_backward re-dispatches on the
            # GLOBAL flag at backward time (a forward-scoped context
            # manager does not protect backward), so the flag must stay
            # off for the life of the process once a CPU DeepSpeech runs.
            # GPU runs use MIOpen and are unaffected.
            torch.backends.mkldnn.enabled = False
        x = self.conv(x)
        b, c, f, t = x.shape
        x = x.reshape(b, c * f, t).permute(2, 0, 1)  # [T, B, C*F]
        for rnn, bn in zip(self.rnns, self.bns):
            out, _ = rnn(x)
            fwd, bwd = out.chunk(2, dim=2)
            x = bn(fwd + bwd)
        return self.fc(x)  # [T, B, n_labels]


def _edit_distance(a, b) -> int:
    """Levenshtein distance over two sequences (words or chars)."""
    if len(a) < len(b):
        a, b = b, a
    prev = list(range(len(b) + 1))
    for i, ca in enumerate(a, 1):
        cur = [i]
        for j, cb in enumerate(b, 1):
            cur.append(min(prev[j] + 1, cur[j - 1] + 1,
                           prev[j - 1] + (ca != cb)))
        prev = cur
    return prev[-1]


class GreedyDecoder:
    """Greedy (best-path) CTC decoder with WER/CER scoring."""

    def __init__(self, labels: str = AN4_LABELS, blank_index: int = 0):
        self.labels = labels
        self.blank = blank_index

    def decode(self, logits: torch.Tensor) -> List[str]:
        # logits: [T, B, n_labels]
        best = logits.argmax(dim=2).t()  # [B, T]
        out = []
        for seq in best:
            chars = []
            prev = self.blank
            for tok in seq.tolist():
                if tok != prev and tok != self.blank:
                    chars.append(self.labels[tok])
                prev = tok
            out.append("".join(chars))
        return out

    def labels_to_string(self, y) -> str:
        """Transcript label indices -> string (for references)."""
        return "".join(self.labels[int(i)] for i in y)

    def wer(self, hyp: str, ref: str) -> float:
        """Word error rate of one utterance (edit distance over words,
        normalised by reference length)."""
        rw = ref.split()
        return _edit_distance(hyp.split(), rw) / max(1, len(rw))

    def cer(self, hyp: str, ref: str) -> float:
        """Character error rate of one utterance."""
        return _edit_distance(hyp, ref) / max(1, len(ref))


def deepspeech_an4(hidden=512):
    return DeepSpeech(hidden=hidden)
