"""Model zoo + ``create_net`` factory.

Reference parity: `upstream:dl_trainer.py` module-level
``create_net(dnn, dataset)``.
"""
from .mnist import MnistFFN, MnistNet  # noqa: F401
from .resnet_cifar import resnet20, resnet56, resnet110  # noqa: F401
from .vgg import VGG, vgg16  # noqa: F401
from .resnet_imagenet import alexnet, resnet50, resnet101  # noqa: F401
from .lstm import LSTMModel, lstm_ptb, repackage_hidden  # noqa: F401
from .deepspeech import (AN4_LABELS, DeepSpeech, GreedyDecoder,  # noqa: F401
                         deepspeech_an4)

_FACTORY = {
    "mnistnet": lambda num_classes=10: MnistNet(num_classes),
    "mnistffn": lambda num_classes=10: MnistFFN(num_classes=num_classes),
    "resnet20": lambda num_classes=10: resnet20(num_classes),
    "resnet56": lambda num_classes=10: resnet56(num_classes),
    "resnet110": lambda num_classes=10: resnet110(num_classes),
    "vgg16": lambda num_classes=10: vgg16(num_classes),
    "resnet50": lambda num_classes=1000: resnet50(num_classes),
    "resnet101": lambda num_classes=1000: resnet101(num_classes),
    "alexnet": lambda num_classes=1000: alexnet(num_classes),
    "lstm": lambda vocab_size=10000, hidden=1500: lstm_ptb(vocab_size, hidden),
    "lstman4": lambda hidden=512: deepspeech_an4(hidden),
}

_DEFAULT_CLASSES = {"mnist": 10, "cifar10": 10, "imagenet": 1000}


def create_net(dnn: str, dataset: str = None, **kwargs):
    """Build a model by name (reference CLI ``--dnn`` values)."""
    dnn = dnn.lower()
    if dnn not in _FACTORY:
        raise ValueError(f"unknown dnn '{dnn}' (have {sorted(_FACTORY)})")
    if dataset in _DEFAULT_CLASSES and dnn not in ("lstm", "lstman4") \
            and "num_classes" not in kwargs:
        kwargs["num_classes"] = _DEFAULT_CLASSES[dataset]
    return _FACTORY[dnn](**kwargs)
