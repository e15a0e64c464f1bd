#!/usr/bin/env python3
"""Preprocessing entry point: walk a wav dir, compute
log-mel spectrograms, save one .mel tensor next to each wav.

    python preprocess.py -c config.yaml -d /path/to/wavs
"""

from __future__ import annotations

import argparse
import glob
import os

import torch

from melgan_amd.config import HParam, merge_default
from melgan_amd.data.dataset import _load_wav
from melgan_amd.data.mel import MelSpectrogram


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-c", "--config", required=True)
    ap.add_argument("-d", "--data_path", required=True)
    args = ap.parse_args()

    hp = merge_default(HParam(args.config))
    mel_fn = MelSpectrogram(
        sr=hp.audio.sampling_rate, n_fft=hp.audio.filter_length,
        hop=hp.audio.hop_length, win=hp.audio.win_length,
        n_mels=hp.audio.n_mel_channels, fmin=hp.audio.mel_fmin,
        fmax=hp.audio.mel_fmax)

    wavs = sorted(glob.glob(os.path.join(args.data_path, "**", "*.wav"),
                            recursive=True))
    print(f"{len(wavs)} wav files")
    for i, path in enumerate(wavs):
        audio, sr = _load_wav(path)
        if sr != hp.audio.sampling_rate:
            raise ValueError(
                f"{path}: sample rate {sr} != config "
                f"{hp.audio.sampling_rate} (resample offline first)")
        mel = mel_fn(audio.unsqueeze(0))[0]  # (n_mels, T)
        torch.save(mel, path.replace(".wav", ".mel"))
        if i % 100 == 0:
            print(f"[{i}/{len(wavs)}] {path} -> mel {tuple(mel.shape)}")


if __name__ == "__main__":
    main()
