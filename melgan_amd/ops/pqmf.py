"""PQMF analysis/synthesis filter bank (cosine-modulated, near-perfect
reconstruction).

Capability contract: SURVEY.md §2.1 "PQMF filter bank" — Kaiser-windowed
prototype lowpass (taps=62, beta=9.0), cosine-modulated analysis H and
synthesis G banks for N = 4 (22.05 kHz) or 8 (44.1 kHz) sub-bands;
`analysis(x)`: filter + stride-N downsample; `synthesis(x)`: xN upsample
+ filter.

MI355X implementation: both directions run as ONE
fused polyphase HIP kernel each — the 63-tap filter lives in LDS and the
down/up-sampling is free in the polyphase indexing (stride-N conv /
gather-form transposed conv).  The CPU path below composes the same math
from torch ops and is the numerics oracle.
"""

from __future__ import annotations

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from melgan_amd.ops import _ext


def design_prototype_filter(taps: int = 62, cutoff_ratio: float = 0.142,
                            beta: float = 9.0) -> np.ndarray:
    """Kaiser-windowed linear-phase lowpass prototype, taps+1 coefficients."""
    from scipy.signal import firwin

    if not (0.0 < cutoff_ratio < 1.0):
        raise ValueError("cutoff_ratio must be in (0, 1)")
    return firwin(taps + 1, cutoff_ratio, window=("kaiser", beta))


# per-band-count near-perfect-reconstruction cutoffs at taps=62/beta=9
# (numerically minimised round-trip error: N=4 -> 0.06% RMS, N=8 -> 0.6%)
_DEFAULT_CUTOFF = {2: 0.265, 4: 0.142, 8: 0.079}


class PQMF(nn.Module):
    def __init__(self, subbands: int = 4, taps: int = 62,
                 cutoff_ratio: float | None = None, beta: float = 9.0):
        super().__init__()
        if cutoff_ratio is None:
            cutoff_ratio = _DEFAULT_CUTOFF.get(subbands, 0.568 / subbands)
        self.subbands = subbands
        self.taps = taps
        h_proto = design_prototype_filter(taps, cutoff_ratio, beta)
        h_analysis = np.zeros((subbands, len(h_proto)))
        h_synthesis = np.zeros((subbands, len(h_proto)))
        n = np.arange(len(h_proto))
        for k in range(subbands):
            c = (2 * k + 1) * (np.pi / (2 * subbands)) * (n - taps / 2)
            phi = (-1) ** k * np.pi / 4
            h_analysis[k] = 2 * h_proto * np.cos(c + phi)
            h_synthesis[k] = 2 * h_proto * np.cos(c - phi)

        # analysis: (N, 1, T) conv filters; synthesis: (N, 1, T) correlation
        self.register_buffer(
            "analysis_filter",
            torch.from_numpy(h_analysis).float().unsqueeze(1), persistent=False)
        self.register_buffer(
            "synthesis_filter",
            torch.from_numpy(h_synthesis).float().unsqueeze(1), persistent=False)
        # x N upsampling scatter filter (family's `updown_filter`)
        updown = torch.zeros((subbands, subbands, subbands))
        for k in range(subbands):
            updown[k, k, 0] = 1.0
        self.register_buffer("updown_filter", updown, persistent=False)
        self.pad = taps // 2

    def analysis(self, x: torch.Tensor) -> torch.Tensor:
        """(B, 1, S) -> (B, N, S/N)"""
        if _ext.use_hip(x):
            from melgan_amd.ops.fused import pqmf_analysis

            return pqmf_analysis(self, x)
        x = F.pad(x, (self.pad, self.pad))
        return F.conv1d(x, self.analysis_filter.to(x.dtype),
                        stride=self.subbands)

    def synthesis(self, x: torch.Tensor) -> torch.Tensor:
        """(B, N, S/N) -> (B, 1, S)"""
        if _ext.use_hip(x):
            from melgan_amd.ops.fused import pqmf_synthesis

            return pqmf_synthesis(self, x)
        n = self.subbands
        x = F.conv_transpose1d(x, self.updown_filter.to(x.dtype) * n, stride=n)
        x = F.pad(x, (self.pad, self.pad))
        # conv1d weight (1, N, taps+1): y = sum_k (up(x_k) * g_k)
        return F.conv1d(x, self.synthesis_filter.to(x.dtype).transpose(0, 1))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """Round trip (analysis then synthesis) — used by the PR test."""
        return self.synthesis(self.analysis(x))
