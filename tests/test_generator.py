import pytest
import torch

from melgan_amd.models import build_generator
from melgan_amd.models.generator import Generator


def test_forward_shape_singleband(tiny_hp):
    g = build_generator(tiny_hp)
    mel = torch.randn(2, 80, 16)
    audio = g(mel)
    assert audio.shape == (2, 1, 16 * 256)
    assert audio.abs().max() <= 1.0  # tanh output


def test_forward_shape_multiband(tiny_mb_hp):
    g = build_generator(tiny_mb_hp)
    mel = torch.randn(2, 80, 16)
    audio = g(mel)
    # 4 sub-bands, each upsampled 64x
    assert audio.shape == (2, 4, 16 * 64)


@pytest.mark.parametrize("ratios,out_ch", [
    ([8, 8, 2, 2], 1), ([8, 4, 2], 4), ([5, 5, 3], 4), ([8, 4, 2], 8)])
def test_all_config_geometries(ratios, out_ch):
    g = Generator(mel_channels=80, out_channels=out_ch, base_channels=32,
                  ratios=ratios)
    t = 8
    up = 1
    for r in ratios:
        up *= r
    y = g(torch.randn(1, 80, t))
    assert y.shape == (1, out_ch, t * up)


def test_remove_weight_norm_identical_output(tiny_hp):
    g = build_generator(tiny_hp)
    g.eval()
    mel = torch.randn(1, 80, 12)
    with torch.no_grad():
        y1 = g(mel)
        g.remove_weight_norm()
        y2 = g(mel)
    assert torch.allclose(y1, y2, atol=1e-5), \
        (y1 - y2).abs().max().item()


def test_weight_norm_param_naming(tiny_hp):
    """Family checkpoint naming contract: *.weight_g / *.weight_v / *.bias"""
    g = build_generator(tiny_hp)
    keys = list(g.state_dict().keys())
    assert any(k.endswith("weight_g") for k in keys)
    assert any(k.endswith("weight_v") for k in keys)
    assert any(k.endswith("bias") for k in keys)
    # no plain .weight before remove_weight_norm
    assert not any(k.endswith(".weight") for k in keys)


def test_inference_helper(tiny_hp):
    g = build_generator(tiny_hp)
    g.eval()
    mel = torch.randn(80, 20)
    audio = g.inference(mel)
    assert audio.shape == (1, 1, 20 * 256)


def test_backward_flows(tiny_mb_hp):
    g = build_generator(tiny_mb_hp)
    mel = torch.randn(2, 80, 8)
    y = g(mel)
    y.pow(2).mean().backward()
    grads = [p.grad for p in g.parameters()]
    assert all(gr is not None for gr in grads)
    assert all(torch.isfinite(gr).all() for gr in grads)


def test_param_count_order():
    """4-band generator should be in the MB-MelGAN ~2M param class."""
    g = Generator(80, 4, 512, [8, 4, 2], 3)
    n = sum(p.numel() for p in g.parameters())
    assert 1e6 < n < 6e6, n


def test_chunked_inference_matches_full(tiny_mb_hp):
    """Overlap-discard streaming == full forward."""
    g = build_generator(tiny_mb_hp)
    g.eval()
    torch.manual_seed(0)
    mel = torch.randn(1, 80, 300)
    with torch.no_grad():
        full = g(mel)
        chunked = g.inference_chunked(mel, chunk_frames=64, overlap=16)
    assert chunked.shape == full.shape
    assert torch.allclose(chunked, full, atol=1e-5), \
        (chunked - full).abs().max().item()


def test_cpu_singleband_forward_golden():
    """Deterministic golden path (SURVEY.md §4 "numerics golden tests"):
    the CPU composite forward is the reference every GPU kernel test
    compares against — if it silently drifts, the whole numerics suite
    drifts with it.  Constants were produced by this exact construction
    on the pinned torch build."""
    torch.manual_seed(42)
    g = Generator(mel_channels=80, out_channels=1, base_channels=32,
                  ratios=[8, 8, 2, 2], n_residual_layers=3).eval()
    mel = torch.randn(1, 80, 10,
                      generator=torch.Generator().manual_seed(7)) * 2 - 6
    with torch.no_grad():
        y = g(mel)
    assert y.shape == (1, 1, 2560)  # 10 frames x 256 upsampling
    expected_first8 = torch.tensor(
        [0.12831259, 0.16710101, 0.08534521, 0.16874576,
         0.11896617, 0.14558543, 0.13494481, 0.14097007])
    assert torch.allclose(y[0, 0, :8], expected_first8, atol=1e-5), \
        y[0, 0, :8].tolist()
    assert abs(float(y.double().sum()) - 348.20916636) < 1e-2
    assert abs(float(y.mean()) - 0.13601921) < 1e-5


@pytest.mark.parametrize("ratios", [[5, 5, 3], [5, 4, 3], [3, 3, 2]])
def test_chunked_inference_odd_ratios(ratios):
    """Overlap-discard must stay EXACT through odd upsample ratios
    (k=2r, output_padding=1 convT geometry — the 24k VCTK config's
    [5,5,3] path)."""
    g = Generator(80, 4, 32, ratios, 3).eval()
    torch.manual_seed(0)
    mel = torch.randn(1, 80, 300)
    with torch.no_grad():
        full = g(mel)
        chunked = g.inference_chunked(mel, chunk_frames=64, overlap=16)
    assert chunked.shape == full.shape
    assert torch.allclose(chunked, full, atol=1e-5)
