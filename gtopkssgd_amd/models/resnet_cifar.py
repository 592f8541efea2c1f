"""CIFAR-10 ResNets (ResNet-20/56/110, option-A identity shortcuts).

Reference parity: `upstream:resnet.py` — the classic
32×32-input ResNet family of He et al. with parameter-free (option A)
downsampling shortcuts.  Written from the paper's architecture, not the
reference's code.
"""
import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.fusedbn import FusedBNAct


class LambdaPad(nn.Module):
    """Option-A shortcut: stride-2 subsample + zero-pad channels."""

    def __init__(self, planes_out: int):
        super().__init__()
        self.planes_out = planes_out

    def forward(self, x):
        out = x[:, :, ::2, ::2]
        pad = self.planes_out - out.size(1)
        out = F.pad(out, (0, 0, 0, 0, pad // 2, pad - pad // 2))
        if out.is_cuda:  # keep the fused-BN residual path applicable
            out = out.contiguous(memory_format=torch.channels_last)
        return out


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes: int, planes: int, stride: int = 1):
        super().__init__()
        self.conv1 = nn.Conv2d(in_planes, planes, 3, stride=stride,
                               padding=1, bias=False)
        self.bn1 = FusedBNAct(planes, relu=True)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=1, padding=1,
                               bias=False)
        self.bn2 = FusedBNAct(planes, relu=True)  # relu after the add
        if stride != 1 or in_planes != planes:
            self.shortcut = LambdaPad(planes)
        else:
            self.shortcut = nn.Identity()

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        return self.bn2(self.conv2(out), residual=self.shortcut(x))


class CifarResNet(nn.Module):
    def __init__(self, num_blocks: int, num_classes: int = 10):
        super().__init__()
        self.in_planes = 16
        self.conv1 = nn.Conv2d(3, 16, 3, stride=1, padding=1, bias=False)
        self.bn1 = FusedBNAct(16, relu=True)
        self.layer1 = self._make_layer(16, num_blocks, 1)
        self.layer2 = self._make_layer(32, num_blocks, 2)
        self.layer3 = self._make_layer(64, num_blocks, 2)
        self.linear = nn.Linear(64, num_classes)

    def _make_layer(self, planes, blocks, stride):
        layers = []
        for s in [stride] + [1] * (blocks - 1):
            layers.append(BasicBlock(self.in_planes, planes, s))
            self.in_planes = planes
        return nn.Sequential(*layers)

    def forward(self, x):
        out = self.bn1(self.conv1(x))
        out = self.layer3(self.layer2(self.layer1(out)))
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.linear(out)


def resnet20(num_classes=10):
    return CifarResNet(3, num_classes)


def resnet56(num_classes=10):
    return CifarResNet(9, num_classes)


def resnet110(num_classes=10):
    return CifarResNet(18, num_classes)
