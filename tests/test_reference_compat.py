"""Reference-family compatibility: a generator built from PLAIN torch
modules wrapped in torch.nn.utils.weight_norm (the reference's exact
parametrisation) must load into our fused modules via
the checkpoint translation layer and produce identical outputs."""

import warnings

import torch
import torch.nn as nn

from melgan_amd.engine.checkpoint import load_flexible
from melgan_amd.models.generator import Generator


def _reference_generator(mel=80, out_ch=4, base=64, ratios=(8, 4, 2),
                         n_res=3):
    """The reference family's composition: torch weight_norm modules."""
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")  # weight_norm deprecation
        wn = nn.utils.weight_norm

        class RefResBlock(nn.Module):
            def __init__(self, c, d):
                super().__init__()
                self.c1 = wn(nn.Conv1d(c, c, 3, dilation=d))
                self.c2 = wn(nn.Conv1d(c, c, 1))
                self.d = d

            def forward(self, x):
                h = nn.functional.leaky_relu(x, 0.2)
                h = nn.functional.pad(h, (self.d, self.d), mode="reflect")
                h = self.c1(h)
                h = nn.functional.leaky_relu(h, 0.2)
                return x + self.c2(h)

        class RefGen(nn.Module):
            def __init__(self):
                super().__init__()
                layers = []
                c = base
                layers.append(wn(nn.Conv1d(mel, c, 7)))
                for r in ratios:
                    pad = r // 2 if r % 2 == 0 else (r + 1) // 2
                    op = 0 if r % 2 == 0 else 1
                    layers.append(wn(nn.ConvTranspose1d(
                        c, c // 2, 2 * r, stride=r, padding=pad,
                        output_padding=op)))
                    c //= 2
                    for i in range(n_res):
                        layers.append(RefResBlock(c, 3 ** i))
                layers.append(wn(nn.Conv1d(c, out_ch, 7)))
                self.net = nn.ModuleList(layers)
                self.ratios = ratios
                self.n_res = n_res

            def forward(self, x):
                it = iter(self.net)
                x = nn.functional.pad(x, (3, 3), mode="reflect")
                x = next(it)(x)
                for r in self.ratios:
                    x = nn.functional.leaky_relu(x, 0.2)
                    x = next(it)(x)
                    for _ in range(self.n_res):
                        x = next(it)(x)
                x = nn.functional.leaky_relu(x, 0.2)
                x = nn.functional.pad(x, (3, 3), mode="reflect")
                x = torch.tanh(next(it)(x))
                return x

        return RefGen()


def test_reference_weightnorm_checkpoint_loads():
    torch.manual_seed(0)
    ref = _reference_generator()
    state = ref.state_dict()
    # reference naming: *.weight_g / *.weight_v / *.bias
    assert any(k.endswith("weight_g") for k in state)
    assert any(k.endswith("weight_v") for k in state)

    ours = Generator(80, 4, 64, [8, 4, 2], 3)
    load_flexible(ours, state)

    mel = torch.randn(1, 80, 16)
    with torch.no_grad():
        y_ref = ref(mel)
        y_ours = ours(mel)
    assert y_ours.shape == y_ref.shape
    assert torch.allclose(y_ours, y_ref, atol=1e-5), \
        (y_ours - y_ref).abs().max().item()


def test_our_checkpoint_loads_into_reference():
    """Round trip the other way: our state dict into the torch model."""
    torch.manual_seed(1)
    ours = Generator(80, 4, 64, [8, 4, 2], 3)
    ref = _reference_generator()
    load_flexible(ref, ours.state_dict())
    mel = torch.randn(1, 80, 12)
    with torch.no_grad():
        assert torch.allclose(ours(mel), ref(mel), atol=1e-5)


def _reference_tower(ndf=16, n_layers=4, df=4):
    """Family tower: torch weight_norm convs, LeakyReLU between."""
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        wn = nn.utils.weight_norm

        class RefTower(nn.Module):
            def __init__(self):
                super().__init__()
                layers = [wn(nn.Conv1d(1, ndf, 15))]
                c, groups = ndf, 1
                for _ in range(n_layers):
                    c_out = min(c * df, 1024)
                    groups = min(groups * df, c_out // 4)
                    layers.append(wn(nn.Conv1d(c, c_out, 41, stride=df,
                                               padding=20, groups=groups)))
                    c = c_out
                layers.append(wn(nn.Conv1d(c, c, 5, padding=2)))
                layers.append(wn(nn.Conv1d(c, 1, 3, padding=1)))
                self.net = nn.ModuleList(layers)

            def forward(self, x):
                feats = []
                x = nn.functional.pad(x, (7, 7), mode="reflect")
                for i, m in enumerate(self.net):
                    x = m(x)
                    if i < len(self.net) - 1:
                        x = nn.functional.leaky_relu(x, 0.2)
                    feats.append(x)
                return feats

        return RefTower()


def test_reference_discriminator_tower_loads():
    """model_d compat: a family-composition tower's weight-norm state
    dict loads into our fused tower and produces identical features."""
    from melgan_amd.models.discriminator import DiscriminatorTower

    torch.manual_seed(0)
    ref = _reference_tower()
    ours = DiscriminatorTower()
    load_flexible(ours, ref.state_dict())
    x = torch.randn(1, 1, 2048)
    with torch.no_grad():
        f_ref = ref(x)
        f_ours = ours(x)
    assert len(f_ref) == len(f_ours)
    for a, b in zip(f_ours, f_ref):
        assert a.shape == b.shape
        assert torch.allclose(a, b, atol=1e-5), (a - b).abs().max().item()
