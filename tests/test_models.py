"""Every model family builds and runs fwd+bwd on tiny inputs (CPU)."""
import pytest
import torch
import torch.nn as nn

from gtopkssgd_amd import models as zoo


@pytest.mark.parametrize("dnn,dataset,shape", [
    ("mnistnet", "mnist", (2, 1, 28, 28)),
    ("mnistffn", "mnist", (2, 1, 28, 28)),
    ("resnet20", "cifar10", (2, 3, 32, 32)),
    ("vgg16", "cifar10", (2, 3, 32, 32)),
])
def test_small_image_models(dnn, dataset, shape):
    net = zoo.create_net(dnn, dataset)
    x = torch.randn(shape)
    out = net(x)
    assert out.shape == (shape[0], 10)
    out.sum().backward()
    assert all(p.grad is not None for p in net.parameters()
               if p.requires_grad)


@pytest.mark.parametrize("dnn", ["resnet50", "alexnet"])
def test_imagenet_models(dnn):
    net = zoo.create_net(dnn, "imagenet")
    x = torch.randn(1, 3, 224, 224)
    out = net(x)
    assert out.shape == (1, 1000)
    out.sum().backward()


def test_resnet50_param_count():
    """ResNet-50 must be the real thing: ~25.56M parameters."""
    net = zoo.resnet50()
    n = sum(p.numel() for p in net.parameters())
    assert abs(n - 25_557_032) < 100_000, n


def test_cifar_resnet_variants():
    for fn, blocks in [(zoo.resnet20, 3), (zoo.resnet56, 9),
                       (zoo.resnet110, 18)]:
        net = fn()
        convs = sum(1 for m in net.modules()
                    if isinstance(m, torch.nn.Conv2d))
        assert convs == 6 * blocks + 1


def test_lstm_lm():
    net = zoo.lstm_ptb(vocab_size=100, hidden=32)
    tokens = torch.randint(0, 100, (5, 3))
    hidden = net.init_hidden(3)
    logits, hidden2 = net(tokens, hidden)
    assert logits.shape == (5, 3, 100)
    logits.sum().backward()
    h = zoo.repackage_hidden(hidden2)
    assert not h[0].requires_grad


def test_deepspeech_ctc():
    net = zoo.deepspeech_an4(hidden=32)
    x = torch.randn(2, 1, 161, 50)
    logits = net(x)
    assert logits.dim() == 3 and logits.size(1) == 2
    assert logits.size(2) == len(zoo.AN4_LABELS)
    # CTC loss runs
    log_probs = logits.log_softmax(2)
    t, b, _ = log_probs.shape
    loss = nn.CTCLoss(blank=0)(log_probs,
                               torch.randint(1, 28, (b, 10)),
                               torch.full((b,), t, dtype=torch.long),
                               torch.full((b,), 10, dtype=torch.long))
    loss.backward()


def test_greedy_decoder():
    dec = zoo.GreedyDecoder()
    t, b, l = 6, 1, len(zoo.AN4_LABELS)
    logits = torch.full((t, b, l), -10.0)
    # spell "AB" with blanks and repeats: A A _ B B _
    a = zoo.AN4_LABELS.index("A")
    bb = zoo.AN4_LABELS.index("B")
    for i, lab in enumerate([a, a, 0, bb, bb, 0]):
        logits[i, 0, lab] = 10.0
    assert dec.decode(logits) == ["AB"]


def test_create_net_rejects_unknown():
    with pytest.raises(ValueError):
        zoo.create_net("nope")
