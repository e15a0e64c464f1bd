"""Multi-band MelGAN generator, MI355X-native.

Architecture (capability contract from SURVEY.md §2.1 "Generator" /
§3.4 — ConvTranspose1d upsampling blocks + dilated residual Conv1d
stacks, weight-norm + LeakyReLU, Tanh output):

    mel (B, 80, T)
    -> refl-pad(3) + wn-Conv1d(80 -> C, k=7)
    -> for r in ratios: [LeakyReLU -> wn-ConvTranspose1d(C -> C/2, k=2r, s=r)
                         -> ResStack(C/2, n_residual_layers, dil 3^i)]
    -> LeakyReLU -> refl-pad(3) + wn-Conv1d(-> out_channels, k=7) -> Tanh
    -> (B, out_channels, T * hop / out_channels)

Every conv is a single fused MI355X op (melgan_amd.ops.wnconv): weight
norm folded, reflection pad folded into addressing, activation in the
kernel epilogue.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from melgan_amd.ops.wnconv import WNConv1d, WNConvTranspose1d


class ResBlock(nn.Module):
    """One dilated residual block: x + f(x),
    f = LReLU -> pad -> wn-Conv1d(k3, dil) -> LReLU -> wn-Conv1d(k1)."""

    def __init__(self, channels: int, dilation: int):
        super().__init__()
        # LeakyReLU fused as the conv's prologue
        self.conv_dil = WNConv1d(channels, channels, 3, dilation=dilation,
                                 pad_mode="reflect", act_before=True)
        self.conv_1x1 = WNConv1d(channels, channels, 1, pad_mode="none",
                                 act_before=True)

    def forward(self, x):
        return x + self.conv_1x1(self.conv_dil(x))

    def remove_weight_norm(self):
        self.conv_dil.remove_weight_norm()
        self.conv_1x1.remove_weight_norm()


class ResStack(nn.Module):
    def __init__(self, channels: int, n_layers: int = 3, dilation_base: int = 3):
        super().__init__()
        self.blocks = nn.ModuleList(
            [ResBlock(channels, dilation_base ** i) for i in range(n_layers)]
        )

    def forward(self, x):
        for b in self.blocks:
            x = b(x)
        return x

    def remove_weight_norm(self):
        for b in self.blocks:
            b.remove_weight_norm()


class Generator(nn.Module):
    """Multi-band MelGAN generator.

    Args mirror the reference YAML `model:` group:
        mel_channels     audio.n_mel_channels (80)
        out_channels     model.out_channels (1 / 4 / 8 sub-bands)
        base_channels    model.base_channels (512)
        ratios           model.generator_ratio (prod*out_channels == hop)
        n_residual_layers model.n_residual_layers (3; 4 for the 44.1 kHz
                          deep-residual config)
    """

    def __init__(self, mel_channels: int = 80, out_channels: int = 4,
                 base_channels: int = 512, ratios=(8, 4, 2),
                 n_residual_layers: int = 3):
        super().__init__()
        self.mel_channels = mel_channels
        self.out_channels = out_channels
        self.ratios = list(ratios)
        self.hop_length = out_channels
        for r in self.ratios:
            self.hop_length *= r

        layers: list[nn.Module] = []
        c = base_channels
        layers.append(WNConv1d(mel_channels, c, 7, pad_mode="reflect"))
        for r in self.ratios:
            layers.append(WNConvTranspose1d(c, c // 2, r, act_before=True))
            c //= 2
            layers.append(ResStack(c, n_residual_layers))
        layers.append(WNConv1d(c, out_channels, 7, pad_mode="reflect",
                               act="tanh", act_before=False, bias=True))
        # final conv's input LeakyReLU is fused as a standalone prologue on
        # the k=7 conv: emulate family's LeakyReLU before the last conv
        self._final_prelrelu = True
        self.layers = nn.ModuleList(layers)

    def forward(self, mel: torch.Tensor) -> torch.Tensor:
        x = mel
        n = len(self.layers)
        for i, layer in enumerate(self.layers):
            if i == n - 1 and self._final_prelrelu:
                x = torch.nn.functional.leaky_relu(x, 0.2)
            x = layer(x)
        return x

    def remove_weight_norm(self):
        for layer in self.layers:
            layer.remove_weight_norm()

    @torch.no_grad()
    def inference_chunked(self, mel: torch.Tensor, chunk_frames: int = 512,
                          overlap: int = 16) -> torch.Tensor:
        """Streaming inference for multi-minute audio:
        process the mel in chunks with `overlap` frames of context on
        each side and stitch by exact overlap-discard (the generator is
        fully convolutional; `overlap` >= its receptive field in frames,
        which is < 10 for every shipped config).  Memory is O(chunk)
        instead of O(utterance)."""
        if mel.dim() == 2:
            mel = mel.unsqueeze(0)
        T = mel.size(2)
        if T <= chunk_frames + 2 * overlap:
            return self.forward(mel)
        hop_sub = self.hop_length // self.out_channels
        outs = []
        for i0 in range(0, T, chunk_frames):
            i1 = min(i0 + chunk_frames, T)
            lo = max(0, i0 - overlap)
            hi = min(T, i1 + overlap)
            y = self.forward(mel[:, :, lo:hi])
            left = (i0 - lo) * hop_sub
            right = (hi - i1) * hop_sub
            outs.append(y[:, :, left:y.size(2) - right])
        return torch.cat(outs, dim=2)

    @torch.no_grad()
    def inference(self, mel: torch.Tensor, pad_frames: int = 10) -> torch.Tensor:
        """Family-style inference helper: pad the mel with
        `pad_frames` frames of silence on each side, run forward, trim the
        corresponding samples from the output."""
        if mel.dim() == 2:
            mel = mel.unsqueeze(0)
        silence = mel.new_full(
            (mel.size(0), mel.size(1), pad_frames), -11.5129251)  # log(1e-5)
        mel = torch.cat([silence, mel, silence], dim=2)
        audio = self.forward(mel)
        hop_sub = self.hop_length // self.out_channels
        trim = pad_frames * hop_sub
        return audio[:, :, trim:-trim]
