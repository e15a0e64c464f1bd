"""Datasets: on-disk mel/wav pairs + synthetic benchmark data.

Capability contract: SURVEY.md §2.1 "Dataset" — `MelFromDisk(hp, train)`
pairs each wav with its precomputed mel, randomly crops aligned
`segment_length`-sample / `segment_length//hop`-frame windows, pads
short clips (`pad_short`), yields (mel[80, T_frames], audio[1, T_samples]).

`SyntheticMelDataset` replaces preprocessing for benchmarking
(BASELINE.json: synthetic mels + random-init weights — there is no
network for real datasets): deterministic per-index random mel/audio
pairs of exactly the configured shapes.
"""

from __future__ import annotations

import glob
import os
import random

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader, Dataset

from melgan_amd.data.mel import MelSpectrogram


def _load_wav(path: str):
    """Minimal 16-bit PCM WAV reader (no soundfile/librosa in the image)."""
    import wave

    import numpy as np

    with wave.open(path, "rb") as w:
        n = w.getnframes()
        sr = w.getframerate()
        sampwidth = w.getsampwidth()
        data = w.readframes(n)
    if sampwidth == 2:
        audio = np.frombuffer(data, dtype=np.int16).astype(np.float32) / 32768.0
    elif sampwidth == 4:
        audio = np.frombuffer(data, dtype=np.int32).astype(np.float32) / 2147483648.0
    else:
        raise ValueError(f"unsupported WAV sample width {sampwidth}")
    if w.getnchannels() > 1:
        audio = audio.reshape(-1, w.getnchannels()).mean(axis=1)
    return torch.from_numpy(audio), sr


class MelFromDisk(Dataset):
    """wav + precomputed mel (.mel / .npy next to the wav)."""

    def __init__(self, hp, train: bool = True):
        self.hp = hp
        self.train = train
        data_dir = hp.data.train_dir if train else hp.data.val_dir
        self.wav_files = sorted(
            glob.glob(os.path.join(data_dir, "**", "*.wav"), recursive=True))
        if not self.wav_files:
            raise FileNotFoundError(f"no .wav files under {data_dir}")
        self.seg_len = hp.audio.segment_length
        self.hop = hp.audio.hop_length
        self.seg_frames = self.seg_len // self.hop
        self.pad_short = hp.audio.pad_short

    def __len__(self):
        return len(self.wav_files)

    def _mel_path(self, wav_path: str) -> str:
        for ext in (".mel", ".mel.npy", ".npy"):
            p = wav_path.replace(".wav", ext)
            if os.path.exists(p):
                return p
        raise FileNotFoundError(
            f"no precomputed mel next to {wav_path}; run preprocess.py")

    def __getitem__(self, idx):
        wav_path = self.wav_files[idx]
        audio, _sr = _load_wav(wav_path)
        mel_path = self._mel_path(wav_path)
        if mel_path.endswith(".npy"):
            import numpy as np

            mel = torch.from_numpy(np.load(mel_path)).float()
        else:
            mel = torch.load(mel_path, map_location="cpu", weights_only=True).float()
        if mel.dim() == 3:
            mel = mel.squeeze(0)

        if audio.numel() < self.seg_len + self.pad_short:
            audio = F.pad(audio, (0, self.seg_len + self.pad_short - audio.numel()))
        audio = audio.unsqueeze(0)

        if self.train:
            max_start = min(mel.size(1) - self.seg_frames - 1,
                            (audio.size(1) - self.seg_len) // self.hop)
            start = random.randint(0, max(0, max_start))
            mel = mel[:, start:start + self.seg_frames]
            audio = audio[:, start * self.hop:start * self.hop + self.seg_len]
            if mel.size(1) < self.seg_frames:
                mel = F.pad(mel, (0, self.seg_frames - mel.size(1)),
                            value=-11.5129251)
            if audio.size(1) < self.seg_len:
                audio = F.pad(audio, (0, self.seg_len - audio.size(1)))
        return mel, audio


class SyntheticMelDataset(Dataset):
    """Deterministic random (mel, audio) pairs of the configured shapes.

    The audio is white noise and the mel is ITS actual log-mel (computed
    once per index on CPU) so the STFT losses see consistent pairs.
    For pure-throughput benchmarking (`quick=True`) both are independent
    random tensors — shape-identical, no CPU mel cost.
    """

    def __init__(self, hp, length: int = 512, quick: bool = True):
        self.hp = hp
        self.length = length
        self.quick = quick
        self.seg_len = hp.audio.segment_length
        self.hop = hp.audio.hop_length
        self.n_mels = hp.audio.n_mel_channels
        self.seg_frames = self.seg_len // self.hop
        if not quick:
            self.mel_fn = MelSpectrogram(
                sr=hp.audio.sampling_rate, n_fft=hp.audio.filter_length,
                hop=hp.audio.hop_length, win=hp.audio.win_length,
                n_mels=self.n_mels, fmin=hp.audio.mel_fmin,
                fmax=hp.audio.mel_fmax)

    def __len__(self):
        return self.length

    def __getitem__(self, idx):
        gen = torch.Generator().manual_seed(1234 + idx)
        audio = torch.rand((1, self.seg_len), generator=gen) * 1.6 - 0.8
        if self.quick:
            mel = torch.randn((self.n_mels, self.seg_frames), generator=gen) \
                * 2.0 - 6.0
        else:
            mel = self.mel_fn(audio)[0, :, :self.seg_frames]
        return mel, audio


def create_dataloader(hp, train: bool = True, synthetic: bool = False,
                      rank: int = 0, world_size: int = 1):
    """Family entry point + DP sharding."""
    if synthetic or (hp.data is None or hp.data.get("train_dir") is None):
        dataset: Dataset = SyntheticMelDataset(hp)
    else:
        dataset = MelFromDisk(hp, train)
    sampler = None
    if world_size > 1:
        sampler = torch.utils.data.distributed.DistributedSampler(
            dataset, num_replicas=world_size, rank=rank, shuffle=train)
    # validation items from disk are FULL-LENGTH uncropped clips of varying
    # size — default_collate can't batch them, so val loaders use batch 1
    # (the family convention); synthetic data is fixed-shape and batches fine
    batch_size = hp.train.batch_size \
        if (train or isinstance(dataset, SyntheticMelDataset)) else 1
    return DataLoader(
        dataset,
        batch_size=batch_size,
        shuffle=(sampler is None and train),
        sampler=sampler,
        num_workers=hp.train.num_workers,
        pin_memory=torch.cuda.is_available(),
        drop_last=train,
        persistent_workers=hp.train.num_workers > 0,
    )
