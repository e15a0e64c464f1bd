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


# ---------------------------------------------------------------------------
# MelFromDisk crop/pad invariants (SURVEY.md §2.1 Dataset: aligned random
# segment crop + pad_short) across clip-length boundaries.
# ---------------------------------------------------------------------------
def _write_corpus(tmp_path, n_samples, hop=256, n_mels=80, sr=22050):
    import wave

    import numpy as np

    t = np.arange(n_samples)
    sig = 0.5 * np.sin(2 * np.pi * 220 * t / sr)
    pcm = np.clip(sig * 32767, -32768, 32767).astype(np.int16)
    p = tmp_path / "clip.wav"
    with wave.open(str(p), "wb") as w:
        w.setnchannels(1)
        w.setsampwidth(2)
        w.setframerate(sr)
        w.writeframes(pcm.tobytes())
    mel = torch.randn(n_mels, n_samples // hop + 1)
    torch.save(mel, str(tmp_path / "clip.mel"))
    return mel


import pytest  # noqa: E402


@pytest.mark.parametrize("n_samples", [
    100,            # far below segment: pad_short path
    4095, 4096,     # segment boundary
    5095, 5096, 5097,  # segment + pad_short boundary
    3 * 4096,       # long clip: real random crop
])
def test_melfromdisk_crop_and_pad(tmp_path, n_samples):
    import random

    import torch.nn.functional as F

    from melgan_amd.config import HParam, merge_default
    from melgan_amd.data.dataset import MelFromDisk, _load_wav

    seg, hop, pad_short = 4096, 256, 1000
    raw = f"""
data: {{train_dir: {tmp_path}, val_dir: {tmp_path}}}
train: {{batch_size: 1, num_workers: 0}}
audio: {{segment_length: {seg}, hop_length: {hop}, pad_short: {pad_short}}}
"""
    hp = merge_default(HParam(raw=raw))
    mel_full = _write_corpus(tmp_path, n_samples, hop=hop)
    ds = MelFromDisk(hp, train=True)
    assert len(ds) == 1

    random.seed(3)
    mel, audio = ds[0]
    frames = seg // hop
    assert mel.shape == (80, frames)
    assert audio.shape == (1, seg)
    assert float(audio.abs().max()) <= 1.0

    # reproduce the crop: same single random.randint drawn from seed 3
    orig, _ = _load_wav(str(tmp_path / "clip.wav"))
    if orig.numel() < seg + pad_short:
        orig = F.pad(orig, (0, seg + pad_short - orig.numel()))
    random.seed(3)
    max_start = min(mel_full.size(1) - frames - 1,
                    (orig.numel() - seg) // hop)
    start = random.randint(0, max(0, max_start))

    # audio window is the original samples at the aligned offset
    assert torch.equal(audio[0], orig[start * hop:start * hop + seg])
    # mel window is the matching frames of the full mel (padded with the
    # log(1e-5) silence value if the clip's mel ran short)
    want = mel_full[:, start:start + frames]
    if want.size(1) < frames:
        want = F.pad(want, (0, frames - want.size(1)), value=-11.5129251)
    assert torch.equal(mel, want)
