"""Audio front-end (data/audio.py): STFT log-spectrogram featurizer,
transcript parsing, waveform-backed datasets (SURVEY.md §2.1 "AN4 audio
stack" — SpectrogramDataset parity)."""
import torch

from gtopkssgd_amd.data.audio import (SpectrogramDataset,
                                      SyntheticAN4Waveforms,
                                      ctc_collate_padded, labels_to_text,
                                      spectrogram, synth_waveform,
                                      text_to_labels)
from gtopkssgd_amd.data.synthetic import make_dataset, make_loader
from gtopkssgd_amd.models.deepspeech import AN4_LABELS, DeepSpeech


def test_spectrogram_shape_161_bins():
    # 16 kHz / 20 ms window -> n_fft=320 -> 161 bins (DeepSpeech default)
    wave = torch.randn(16000)  # 1 s
    x = spectrogram(wave)
    assert x.shape[0] == 1 and x.shape[1] == 161
    # 10 ms hop, center=True: ~101 frames for 1 s
    assert 95 <= x.shape[2] <= 105


def test_spectrogram_normalized_and_deterministic():
    wave = torch.sin(torch.arange(8000, dtype=torch.float32) * 0.3)
    a = spectrogram(wave)
    b = spectrogram(wave)
    assert torch.equal(a, b)
    assert abs(a.mean().item()) < 1e-3
    assert abs(a.std().item() - 1.0) < 1e-2


def test_text_labels_roundtrip():
    y = text_to_labels("HELLO WORLD")
    assert labels_to_text(y.tolist()) == "HELLO WORLD"
    # lowercase folds to upper; unknown chars dropped
    assert labels_to_text(text_to_labels("a-b?c").tolist()) == "ABC"


def test_synth_waveform_deterministic_and_sized():
    y = torch.tensor([3, 5, 7])
    w1 = synth_waveform(y, seed=9)
    w2 = synth_waveform(y, seed=9)
    assert torch.equal(w1, w2)
    assert w1.numel() == 3 * int(16000 * 0.120)
    assert not torch.equal(w1, synth_waveform(y, seed=10))


def test_spectrogram_dataset_from_waveforms():
    src = [(torch.randn(8000), "AB C"), (torch.randn(12000), "HELLO")]
    ds = SpectrogramDataset(src)
    assert len(ds) == 2
    x0, y0 = ds[0]
    x1, y1 = ds[1]
    assert x0.shape[1] == 161 and x1.shape[1] == 161
    assert x1.shape[2] > x0.shape[2]  # longer audio, more frames
    assert labels_to_text(y0.tolist()) == "AB C"


def test_ctc_collate_pads_time_axis():
    src = [(torch.randn(8000), "AB"), (torch.randn(16000), "CDE")]
    ds = SpectrogramDataset(src)
    x, ys = ctc_collate_padded([ds[0], ds[1]])
    assert x.shape[0] == 2 and x.shape[1] == 1 and x.shape[2] == 161
    assert x.shape[3] == max(ds[0][0].shape[2], ds[1][0].shape[2])
    assert [len(y) for y in ys] == [2, 3]


def test_waveform_dataset_feeds_deepspeech():
    ds = make_dataset("an4wav", n_samples=4, seed=1, max_transcript=4)
    assert isinstance(ds, SyntheticAN4Waveforms)
    loader = make_loader(ds, batch_size=2, num_workers=0)
    x, ys = next(iter(loader))
    model = DeepSpeech(freq_bins=161, hidden=16, num_layers=1)
    logits = model(x)
    assert logits.shape[1] == 2 and logits.shape[2] == len(AN4_LABELS)
    # CTC feasibility: enough output frames for every transcript
    assert logits.shape[0] >= max(len(y) for y in ys)


def test_trainer_runs_on_waveform_dataset():
    """End-to-end: waveform STFT front-end through DLTrainer's CTC
    train+eval path (dataset='an4wav')."""
    from gtopkssgd_amd.trainer import DLTrainer
    trainer = DLTrainer(dnn="lstman4", dataset="an4wav", batch_size=2,
                        device="cpu", amp=False, num_workers=0,
                        n_samples=4, model_kwargs={"hidden": 16})
    trainer.train(1)
    metrics = trainer.test(0)
    assert set(metrics) == {"ctc_loss", "wer", "cer"}


def test_label_tones_are_distinguishable_in_spectrogram():
    """Different labels must produce different spectral signatures —
    the property that makes the synthetic CTC task learnable from
    real STFT features."""
    a = spectrogram(synth_waveform(torch.tensor([2] * 4), seed=0))
    b = spectrogram(synth_waveform(torch.tensor([9] * 4), seed=0))
    prof_a = a.mean(dim=2).squeeze(0)
    prof_b = b.mean(dim=2).squeeze(0)
    cos = torch.nn.functional.cosine_similarity(prof_a, prof_b, dim=0)
    assert cos < 0.95, cos
