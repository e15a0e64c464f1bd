import torch

from melgan_amd.data import MelSpectrogram, SyntheticMelDataset, create_dataloader
from melgan_amd.data.mel import mel_filterbank


def test_synthetic_shapes(tiny_hp):
    ds = SyntheticMelDataset(tiny_hp, length=8)
    mel, audio = ds[0]
    frames = tiny_hp.audio.segment_length // tiny_hp.audio.hop_length
    assert mel.shape == (80, frames)
    assert audio.shape == (1, tiny_hp.audio.segment_length)


def test_synthetic_deterministic(tiny_hp):
    ds = SyntheticMelDataset(tiny_hp, length=8)
    m1, a1 = ds[3]
    m2, a2 = ds[3]
    assert torch.equal(m1, m2) and torch.equal(a1, a2)
    m3, _ = ds[4]
    assert not torch.equal(m1, m3)


def test_dataloader(tiny_hp):
    dl = create_dataloader(tiny_hp, train=True, synthetic=True)
    mel, audio = next(iter(dl))
    assert mel.shape[0] == tiny_hp.train.batch_size
    assert audio.shape[2] == tiny_hp.audio.segment_length


def test_mel_filterbank_matches_librosa_shape():
    fb = mel_filterbank(22050, 1024, 80, 0.0, 8000.0)
    assert fb.shape == (80, 513)
    assert (fb >= 0).all()
    assert (fb.sum(axis=1) > 0).all()  # every filter non-empty


def test_melspectrogram(tiny_hp):
    mel_fn = MelSpectrogram()
    wav = torch.randn(2, 22050)
    mel = mel_fn(wav)
    assert mel.shape[0] == 2 and mel.shape[1] == 80
    assert torch.isfinite(mel).all()
    assert mel.min() >= torch.log(torch.tensor(1e-5)) - 1e-4
