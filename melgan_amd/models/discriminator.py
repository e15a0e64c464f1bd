"""Multi-scale discriminator, MI355X-native.

Capability contract (SURVEY.md §2.1 "Sub-band discriminator" /
"Multi-scale discriminator", §3.5): three identical towers applied to
the audio at 1x, 2x and 4x average-pooled scales; each tower is
    wn-Conv1d(1, ndf, k=15)                       -> LReLU
    4x wn-Conv1d(k=41, stride=4, groups 4..256)   -> LReLU
    wn-Conv1d(ndf*64, ndf*64, k=5)                -> LReLU
    wn-Conv1d(ndf*64, 1, k=3)
returning ALL intermediate feature maps plus the final score (for the
feature-matching loss).

The strided/grouped k=41 convs are the hot kernels;
on MI355X they run as fused weight-norm+LeakyReLU grouped convs with the
41-tap window staged through LDS.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from melgan_amd.ops.wnconv import WNConv1d


class DiscriminatorTower(nn.Module):
    def __init__(self, ndf: int = 16, n_layers: int = 4,
                 downsamp_factor: int = 4, max_channels: int = 1024):
        super().__init__()
        layers = []
        layers.append(WNConv1d(1, ndf, 15, pad_mode="reflect",
                               act="leaky", act_slope=0.2))
        c = ndf
        groups = 1
        for _ in range(n_layers):
            c_out = min(c * downsamp_factor, max_channels)
            groups = min(groups * downsamp_factor, c_out // 4)
            layers.append(WNConv1d(c, c_out, 41, stride=downsamp_factor,
                                   groups=groups, pad_mode="zero",
                                   act="leaky", act_slope=0.2))
            c = c_out
        layers.append(WNConv1d(c, c, 5, pad_mode="zero",
                               act="leaky", act_slope=0.2))
        layers.append(WNConv1d(c, 1, 3, pad_mode="zero", act=None))
        self.layers = nn.ModuleList(layers)

    def forward(self, x: torch.Tensor):
        feats = []
        for layer in self.layers:
            x = layer(x)
            feats.append(x)
        return feats  # [..., score]; score == feats[-1]


class MultiScaleDiscriminator(nn.Module):
    """num_D towers over avg-pooled scales (identity, /2, /4)."""

    def __init__(self, num_D: int = 3, ndf: int = 16, n_layers: int = 4,
                 downsamp_factor: int = 4):
        super().__init__()
        self.towers = nn.ModuleList(
            [DiscriminatorTower(ndf, n_layers, downsamp_factor)
             for _ in range(num_D)]
        )

    def forward(self, audio: torch.Tensor):
        """audio: (B, 1, S) -> list over scales of [feat..., score]."""
        from melgan_amd.ops import _ext

        results = []
        x = audio
        use_hip = _ext.use_hip(audio) and _ext.ext() is not None and \
            hasattr(_ext.ext(), "avg_pool1d_421")
        for i, tower in enumerate(self.towers):
            if i > 0:
                if use_hip:
                    from melgan_amd.ops.fused import fused_avg_pool421

                    x = fused_avg_pool421(x)
                else:
                    x = F.avg_pool1d(x, kernel_size=4, stride=2, padding=1,
                                     count_include_pad=False)
            results.append(tower(x))
        return results
