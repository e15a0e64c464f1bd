"""Mel-spectrogram extraction (Tacotron-style, librosa-free).

Capability contract: SURVEY.md §2.1 "Mel/audio DSP" — TacotronSTFT:
filter_length=1024, hop=256, win=1024, n_mels=80, fmin=0, fmax=8000,
sr=22050; dynamic-range compression log(clamp(x, 1e-5)).

The Slaney-scale mel filterbank is built locally (numpy) so the
offline image needs no librosa; values match librosa.filters.mel
(htk=False, norm='slaney') to float precision.
"""

from __future__ import annotations

import numpy as np
import torch
import torch.nn as nn


def _hz_to_mel(f):
    """Slaney mel scale (librosa default)."""
    f = np.asanyarray(f, dtype=np.float64)
    f_sp = 200.0 / 3
    mels = f / f_sp
    min_log_hz = 1000.0
    min_log_mel = min_log_hz / f_sp
    logstep = np.log(6.4) / 27.0
    above = f >= min_log_hz
    mels = np.where(above, min_log_mel + np.log(np.maximum(f, 1e-10) / min_log_hz) / logstep, mels)
    return mels


def _mel_to_hz(m):
    m = np.asanyarray(m, dtype=np.float64)
    f_sp = 200.0 / 3
    freqs = f_sp * m
    min_log_hz = 1000.0
    min_log_mel = min_log_hz / f_sp
    logstep = np.log(6.4) / 27.0
    above = m >= min_log_mel
    freqs = np.where(above, min_log_hz * np.exp(logstep * (m - min_log_mel)), freqs)
    return freqs


def mel_filterbank(sr: int, n_fft: int, n_mels: int,
                   fmin: float = 0.0, fmax: float | None = None) -> np.ndarray:
    """Slaney-normalised triangular mel filterbank, (n_mels, n_fft//2+1)."""
    if fmax is None:
        fmax = sr / 2.0
    fft_freqs = np.linspace(0, sr / 2.0, n_fft // 2 + 1)
    mel_pts = np.linspace(_hz_to_mel(fmin), _hz_to_mel(fmax), n_mels + 2)
    hz_pts = _mel_to_hz(mel_pts)
    fb = np.zeros((n_mels, len(fft_freqs)))
    fdiff = np.diff(hz_pts)
    ramps = hz_pts.reshape(-1, 1) - fft_freqs.reshape(1, -1)
    for i in range(n_mels):
        lower = -ramps[i] / fdiff[i]
        upper = ramps[i + 2] / fdiff[i + 1]
        fb[i] = np.maximum(0, np.minimum(lower, upper))
    # slaney norm
    enorm = 2.0 / (hz_pts[2:n_mels + 2] - hz_pts[:n_mels])
    fb *= enorm[:, None]
    return fb


def dynamic_range_compression(x: torch.Tensor, clip_val: float = 1e-5):
    return torch.log(torch.clamp(x, min=clip_val))


class MelSpectrogram(nn.Module):
    """wav (B, S) in [-1, 1] -> log-mel (B, n_mels, frames)."""

    def __init__(self, sr: int = 22050, n_fft: int = 1024, hop: int = 256,
                 win: int = 1024, n_mels: int = 80, fmin: float = 0.0,
                 fmax: float | None = 8000.0):
        super().__init__()
        self.n_fft = n_fft
        self.hop = hop
        self.win = win
        fb = mel_filterbank(sr, n_fft, n_mels, fmin, fmax)
        self.register_buffer("mel_basis", torch.from_numpy(fb).float(),
                             persistent=False)
        self.register_buffer("window", torch.hann_window(win),
                             persistent=False)

    @torch.no_grad()
    def forward(self, wav: torch.Tensor) -> torch.Tensor:
        if wav.dim() == 3:
            wav = wav.squeeze(1)
        spec = torch.stft(wav.float(), self.n_fft, self.hop, self.win,
                          window=self.window, center=True,
                          pad_mode="reflect", return_complex=True)
        mag = spec.abs()  # (B, n_fft//2+1, frames)
        mel = torch.matmul(self.mel_basis, mag)
        return dynamic_range_compression(mel)
