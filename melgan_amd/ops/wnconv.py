"""Weight-normalised 1-D convolution modules with fused MI355X kernels.

The reference family wraps ``nn.Conv1d`` / ``nn.ConvTranspose1d`` in
``torch.nn.utils.weight_norm`` and applies ``LeakyReLU`` / ``Tanh``
separately (SURVEY.md §2.1 Generator / §3.4).  On MI355X that lowers to
5+ kernel launches and 3 extra HBM round-trips per conv.  Here each conv
is ONE fused op: weight-norm is folded (w = g·v/‖v‖ computed per step on
GPU), reflection padding is folded into the address calculation, and the
activation runs in the conv epilogue.

Parameters are registered under the family's names (``weight_g``,
``weight_v``, ``bias``) so generator checkpoints in the reference's
weight-norm naming load unchanged (SURVEY.md §5.4 checkpoint compat).

Dispatch: on ROCm GPUs the hand-written gfx950 kernels run (loudly
required); on CPU a composite PyTorch path with identical numerics
serves as the test oracle.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from melgan_amd.ops import _ext

ACT_NONE = 0
ACT_LEAKY = 1
ACT_TANH = 2

_ACT_FROM_STR = {None: ACT_NONE, "none": ACT_NONE, "leaky": ACT_LEAKY, "tanh": ACT_TANH}


def weight_norm_weight(v: torch.Tensor, g: torch.Tensor) -> torch.Tensor:
    """w = g * v / ||v||, norm over all dims except dim 0 (torch weight_norm dim=0)."""
    norm = v.flatten(1).norm(dim=1).clamp_min(1e-12)
    shape = (-1,) + (1,) * (v.dim() - 1)
    return v * (g.flatten() / norm).reshape(shape)


def _apply_act(y: torch.Tensor, act: int, slope: float) -> torch.Tensor:
    if act == ACT_LEAKY:
        return F.leaky_relu(y, slope)
    if act == ACT_TANH:
        return torch.tanh(y)
    return y


class WNConv1d(nn.Module):
    """Fused weight-norm Conv1d (+pre-pad, +activation epilogue).

    Equivalent reference composition:
        pad -> weight_norm(nn.Conv1d(...)) -> LeakyReLU/Tanh
    """

    def __init__(
        self,
        in_channels: int,
        out_channels: int,
        kernel_size: int,
        stride: int = 1,
        dilation: int = 1,
        groups: int = 1,
        pad_mode: str = "reflect",  # "reflect" | "zero" | "none"
        act: str | None = None,
        act_slope: float = 0.2,
        act_before: bool = False,  # activation applied to the INPUT (family
        # ordering inside upsample blocks is LeakyReLU->conv; we fuse it as a
        # prologue instead of a separate kernel)
        bias: bool = True,
    ):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = kernel_size
        self.stride = stride
        self.dilation = dilation
        self.groups = groups
        self.pad_mode = pad_mode
        if pad_mode == "none":
            self.pad = 0
        else:
            self.pad = (kernel_size - 1) // 2 * dilation
        self.act = _ACT_FROM_STR[act]
        self.act_slope = act_slope
        self.act_before = act_before

        v = torch.empty(out_channels, in_channels // groups, kernel_size)
        # Kaiming-uniform, matching nn.Conv1d's default init of `weight`
        nn.init.kaiming_uniform_(v, a=math.sqrt(5))
        # split into weight-norm parametrisation exactly like
        # torch.nn.utils.weight_norm(conv): g = ||w||, v = w
        with torch.no_grad():
            g = v.flatten(1).norm(dim=1).reshape(-1, 1, 1)
        self.weight_v = nn.Parameter(v)
        self.weight_g = nn.Parameter(g)
        if bias:
            fan_in = in_channels // groups * kernel_size
            bound = 1.0 / math.sqrt(fan_in)
            b = torch.empty(out_channels).uniform_(-bound, bound)
            self.bias = nn.Parameter(b)
        else:
            self.register_parameter("bias", None)
        self._removed = False  # after remove_weight_norm()

    # -- weight-norm management ------------------------------------------
    def current_weight(self) -> torch.Tensor:
        if self._removed:
            return self.weight
        return weight_norm_weight(self.weight_v, self.weight_g)

    def remove_weight_norm(self):
        """Fold g·v/‖v‖ into a plain `weight` (inference fast path)."""
        if self._removed:
            return
        w = self.current_weight().detach()
        del self._parameters["weight_v"]
        del self._parameters["weight_g"]
        self.weight = nn.Parameter(w)
        self._removed = True

    # -- forward ----------------------------------------------------------
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if _ext.use_hip(x):
            from melgan_amd.ops.fused import fused_conv1d

            return fused_conv1d(self, x)
        return self._composite(x)

    def _composite(self, x: torch.Tensor) -> torch.Tensor:
        if self.act_before:
            x = F.leaky_relu(x, self.act_slope)
        if self.pad > 0:
            mode = "reflect" if self.pad_mode == "reflect" else "constant"
            x = F.pad(x, (self.pad, self.pad), mode=mode)
        w = self.current_weight()
        y = F.conv1d(x, w, self.bias, stride=self.stride,
                     dilation=self.dilation, groups=self.groups)
        if not self.act_before:
            y = _apply_act(y, self.act, self.act_slope)
        return y

    def extra_repr(self):
        return (f"{self.in_channels}, {self.out_channels}, k={self.kernel_size}, "
                f"s={self.stride}, d={self.dilation}, g={self.groups}, "
                f"pad={self.pad_mode}, act={self.act}{'<' if self.act_before else ''}")


class WNConvTranspose1d(nn.Module):
    """Fused weight-norm ConvTranspose1d with LeakyReLU prologue.

    Reference composition (SURVEY.md §3.4 upsample block):
        LeakyReLU -> weight_norm(nn.ConvTranspose1d(C, C/2, k=2r, stride=r))

    The MI355X kernel is written in gather form (output-stationary): each
    output sample t gathers from input positions (t+pad-k')/r — no atomics
    (SURVEY.md §2.2 K2).  Geometry gives exact r× upsampling:
        even r: k=2r, pad=r//2,   output_padding=0
        odd  r: k=2r, pad=(r+1)//2, output_padding=1
    """

    def __init__(self, in_channels: int, out_channels: int, ratio: int,
                 act_before: bool = True, act_slope: float = 0.2):
        super().__init__()
        r = ratio
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.ratio = r
        self.kernel_size = 2 * r
        self.stride = r
        if r % 2 == 0:
            self.padding = r // 2
            self.output_padding = 0
        else:
            self.padding = (r + 1) // 2
            self.output_padding = 1
        self.act_before = act_before
        self.act_slope = act_slope

        # torch ConvTranspose1d weight layout: (C_in, C_out, k); weight_norm dim=0
        v = torch.empty(in_channels, out_channels, self.kernel_size)
        nn.init.kaiming_uniform_(v, a=math.sqrt(5))
        with torch.no_grad():
            g = v.flatten(1).norm(dim=1).reshape(-1, 1, 1)
        self.weight_v = nn.Parameter(v)
        self.weight_g = nn.Parameter(g)
        fan_in = in_channels * self.kernel_size
        bound = 1.0 / math.sqrt(fan_in)
        self.bias = nn.Parameter(torch.empty(out_channels).uniform_(-bound, bound))
        self._removed = False

    def current_weight(self) -> torch.Tensor:
        if self._removed:
            return self.weight
        return weight_norm_weight(self.weight_v, self.weight_g)

    def remove_weight_norm(self):
        if self._removed:
            return
        w = self.current_weight().detach()
        del self._parameters["weight_v"]
        del self._parameters["weight_g"]
        self.weight = nn.Parameter(w)
        self._removed = True

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if _ext.use_hip(x):
            from melgan_amd.ops.fused import fused_conv_transpose1d

            return fused_conv_transpose1d(self, x)
        return self._composite(x)

    def _composite(self, x: torch.Tensor) -> torch.Tensor:
        if self.act_before:
            x = F.leaky_relu(x, self.act_slope)
        w = self.current_weight()
        return F.conv_transpose1d(
            x, w, self.bias, stride=self.stride, padding=self.padding,
            output_padding=self.output_padding)

    def extra_repr(self):
        return (f"{self.in_channels}, {self.out_channels}, r={self.ratio} "
                f"(k={self.kernel_size}, s={self.stride}, p={self.padding}, "
                f"op={self.output_padding})")
