import torch

from melgan_amd.ops.stft_loss import (
    FULLBAND_RESOLUTIONS,
    MultiResolutionSTFTLoss,
    STFTLoss,
    SUBBAND_RESOLUTIONS,
)


def test_zero_for_identical():
    loss = MultiResolutionSTFTLoss()
    x = torch.randn(2, 4096)
    sc, mag = loss(x, x)
    assert float(sc) < 1e-6
    assert float(mag) < 1e-6


def test_positive_for_different():
    torch.manual_seed(0)
    loss = MultiResolutionSTFTLoss()
    x = torch.randn(2, 4096)
    y = torch.randn(2, 4096)
    sc, mag = loss(x, y)
    assert float(sc) > 0.1
    assert float(mag) > 0.1


def test_matches_manual_composition():
    """STFTLoss == the SC/log-mag formulas on torch.stft magnitudes."""
    torch.manual_seed(1)
    f, h, w = 512, 128, 512
    loss = STFTLoss(f, h, w)
    x = torch.randn(1, 4096)
    y = torch.randn(1, 4096)
    sc, mag = loss(x, y)

    win = torch.hann_window(w)
    Sx = torch.stft(x, f, h, w, window=win, return_complex=True,
                    center=True, pad_mode="reflect").abs().clamp_min(1e-7)
    Sy = torch.stft(y, f, h, w, window=win, return_complex=True,
                    center=True, pad_mode="reflect").abs().clamp_min(1e-7)
    sc_ref = torch.norm(Sy - Sx, p="fro") / torch.norm(Sy, p="fro")
    mag_ref = (Sx.log() - Sy.log()).abs().mean()
    assert torch.allclose(sc, sc_ref, atol=1e-5)
    assert torch.allclose(mag, mag_ref, atol=1e-5)


def test_resolution_sets():
    assert FULLBAND_RESOLUTIONS == [(1024, 120, 600), (2048, 240, 1200),
                                    (512, 50, 240)]
    assert SUBBAND_RESOLUTIONS == [(384, 30, 150), (683, 60, 300),
                                   (171, 10, 60)]


def test_multichannel_folding():
    loss = MultiResolutionSTFTLoss(subband=True)
    x = torch.randn(2, 4, 1024)
    y = torch.randn(2, 4, 1024)
    sc, mag = loss(x, y)
    assert torch.isfinite(sc) and torch.isfinite(mag)


def test_gradients():
    loss = MultiResolutionSTFTLoss()
    x = torch.randn(1, 4096, requires_grad=True)
    y = torch.randn(1, 4096)
    sc, mag = loss(x, y)
    (sc + mag).backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()


def test_fp32_under_bf16_inputs():
    loss = MultiResolutionSTFTLoss()
    x = torch.randn(1, 4096, dtype=torch.bfloat16)
    y = torch.randn(1, 4096, dtype=torch.bfloat16)
    sc, mag = loss(x, y)
    assert sc.dtype == torch.float32 and mag.dtype == torch.float32
