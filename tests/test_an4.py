"""AN4 CTC path: greedy decoder WER/CER metrics, trainer eval, and a
convergence sanity run on the learnable synthetic CTC task (SURVEY.md
§2.1 "AN4 audio stack")."""
import torch
import torch.nn as nn

import gtopkssgd_amd as gt
from gtopkssgd_amd.data.synthetic import SyntheticAN4, make_loader
from gtopkssgd_amd.models.deepspeech import (AN4_LABELS, DeepSpeech,
                                             GreedyDecoder, _edit_distance)


def test_edit_distance():
    assert _edit_distance("", "") == 0
    assert _edit_distance("abc", "abc") == 0
    assert _edit_distance("abc", "axc") == 1
    assert _edit_distance("abc", "") == 3
    assert _edit_distance(["a", "b"], ["b"]) == 1
    assert _edit_distance("kitten", "sitting") == 3


def test_wer_cer():
    d = GreedyDecoder()
    assert d.wer("HELLO WORLD", "HELLO WORLD") == 0.0
    assert d.cer("HELLO", "HELLO") == 0.0
    assert d.wer("HELLO THERE", "HELLO WORLD") == 0.5
    assert d.wer("", "A B") == 1.0
    assert abs(d.cer("HALLO", "HELLO") - 0.2) < 1e-9


def test_decoder_collapses_repeats_and_blanks():
    d = GreedyDecoder()
    # T=6, B=1: blank A A blank B B -> "AB"
    tokens = [0, 2, 2, 0, 3, 3]  # labels: _ ' A B ...
    logits = torch.zeros(6, 1, len(AN4_LABELS))
    for t, tok in enumerate(tokens):
        logits[t, 0, tok] = 10.0
    assert d.decode(logits) == ["AB"]


def test_labels_to_string_roundtrip():
    d = GreedyDecoder()
    y = torch.tensor([2, 3, 4, 28])  # A B C space... (28 = ' ')
    s = d.labels_to_string(y)
    assert s == "ABC "


def test_trainer_ctc_test_reports_wer_cer():
    from gtopkssgd_amd.trainer import DLTrainer
    trainer = DLTrainer(dnn="lstman4", dataset="an4", batch_size=4,
                        device="cpu", amp=False, num_workers=0,
                        n_samples=16,
                        model_kwargs={"hidden": 32})
    metrics = trainer.test(0)
    assert set(metrics) == {"ctc_loss", "wer", "cer"}
    assert metrics["ctc_loss"] > 0
    # untrained model on random transcripts: error rates near 1
    assert 0.0 <= metrics["cer"] <= 2.0
    assert 0.0 <= metrics["wer"] <= 2.0


def test_ctc_convergence_learnable_task():
    """gTop-k rho=0.05 on the learnable synthetic CTC task: loss must
    drop and CER must improve over the untrained model (the reference's
    convergence-methodology analogue for AN4)."""
    torch.manual_seed(3)
    n_labels = 6
    model = DeepSpeech(freq_bins=20, hidden=48, num_layers=1,
                       n_labels=n_labels)
    inner = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    opt = gt.DistributedOptimizer(inner, model.named_parameters(),
                                  compression=gt.compressors["gtopk"](0.05))
    crit = nn.CTCLoss(blank=0, zero_infinity=True)
    data = SyntheticAN4(freq_bins=20, time_steps=60, max_transcript=5,
                        n_labels=n_labels, n_samples=64, seed=5,
                        learnable=True)
    loader = make_loader(data, 8, num_workers=0, seed=5)
    decoder = GreedyDecoder(labels=AN4_LABELS[:n_labels])

    def eval_cer():
        model.eval()
        cer, n = 0.0, 0
        with torch.no_grad():
            for x, ys in loader:
                logits = model(x)
                for hyp, y in zip(decoder.decode(logits), ys):
                    cer += decoder.cer(hyp, decoder.labels_to_string(y))
                    n += 1
        return cer / n

    cer0 = eval_cer()
    model.train()
    losses = []
    for epoch in range(6):
        for x, ys in loader:
            logits = model(x)
            log_probs = logits.log_softmax(2)
            t, b, _ = log_probs.shape
            input_lengths = torch.full((b,), t, dtype=torch.long)
            target_lengths = torch.tensor([len(y) for y in ys])
            targets = torch.cat(list(ys))
            opt.zero_grad()
            loss = crit(log_probs, targets, input_lengths, target_lengths)
            loss.backward()
            opt.step()
            losses.append(loss.item())
    first = sum(losses[:4]) / 4
    last = sum(losses[-4:]) / 4
    assert last < 0.7 * first, (first, last)
    cer1 = eval_cer()
    assert cer1 < cer0, (cer0, cer1)
