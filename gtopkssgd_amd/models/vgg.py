"""VGG for CIFAR-10 (cfg-driven conv stacks).

Reference parity: `upstream:vgg.py` — VGG('VGG16') with
the standard cfg dict; written from the VGG paper's layer spec.
"""
import torch.nn as nn

from ..ops.fusedbn import FusedBNAct

cfg = {
    "VGG11": [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "VGG13": [64, 64, "M", 128, 128, "M", 256, 256, "M", 512, 512, "M",
              512, 512, "M"],
    "VGG16": [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
              512, 512, 512, "M", 512, 512, 512, "M"],
    "VGG19": [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M",
              512, 512, 512, 512, "M", 512, 512, 512, 512, "M"],
}


class VGG(nn.Module):
    def __init__(self, vgg_name: str = "VGG16", num_classes: int = 10):
        super().__init__()
        self.features = self._make_layers(cfg[vgg_name])
        self.classifier = nn.Linear(512, num_classes)

    def _make_layers(self, layer_cfg):
        layers = []
        in_ch = 3
        for x in layer_cfg:
            if x == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers += [nn.Conv2d(in_ch, x, 3, padding=1, bias=False),
                           FusedBNAct(x, relu=True)]
                in_ch = x
        layers.append(nn.AvgPool2d(1, 1))
        return nn.Sequential(*layers)

    def forward(self, x):
        out = self.features(x).flatten(1)
        return self.classifier(out)


def vgg16(num_classes=10):
    return VGG("VGG16", num_classes)
