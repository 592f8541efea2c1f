"""MNIST models: small CNN and 2-layer FFN.

Reference parity: `upstream:dl_trainer.py`/`mnistnet.py` (SURVEY.md
§2.1 "MNIST nets").  The FFN is BASELINE.json config 1 (plumbing test,
CPU/gloo world=2).
"""
import torch.nn as nn
import torch.nn.functional as F


class MnistNet(nn.Module):
    """Classic LeNet-style MNIST CNN."""

    def __init__(self, num_classes: int = 10):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 32, 5, padding=2)
        self.conv2 = nn.Conv2d(32, 64, 5, padding=2)
        self.fc1 = nn.Linear(64 * 7 * 7, 512)
        self.fc2 = nn.Linear(512, num_classes)

    def forward(self, x):
        x = F.max_pool2d(F.relu(self.conv1(x)), 2)
        x = F.max_pool2d(F.relu(self.conv2(x)), 2)
        x = x.flatten(1)
        return self.fc2(F.relu(self.fc1(x)))


class MnistFFN(nn.Module):
    """2-layer fully-connected net (BASELINE config 1)."""

    def __init__(self, hidden: int = 512, num_classes: int = 10):
        super().__init__()
        self.fc1 = nn.Linear(28 * 28, hidden)
        self.fc2 = nn.Linear(hidden, num_classes)

    def forward(self, x):
        x = x.flatten(1)
        return self.fc2(F.relu(self.fc1(x)))
