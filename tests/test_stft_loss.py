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


def test_family_dict_resolutions_schema():
    """The reference family writes train.stft_loss_params as a dict of
    lists (fft_sizes/hop_sizes/win_lengths) — both that and the list of
    (fft, hop, win) triples must construct the same loss."""
    from melgan_amd.config import HParam, merge_default
    from melgan_amd.ops.stft_loss import MultiResolutionSTFTLoss

    raw = """
train:
  stft_loss_params:
    fft_sizes: [1024, 2048, 512]
    hop_sizes: [120, 240, 50]
    win_lengths: [600, 1200, 240]
audio: {hop_length: 256}
model: {out_channels: 4, base_channels: 32, generator_ratio: [8, 4, 2]}
"""
    hp = merge_default(HParam(raw=raw))
    m_dict = MultiResolutionSTFTLoss(resolutions=hp.train.stft_loss_params)
    m_list = MultiResolutionSTFTLoss(
        resolutions=[(1024, 120, 600), (2048, 240, 1200), (512, 50, 240)])
    got = [(l.fft_size, l.hop_size, l.win_length) for l in m_dict.losses]
    want = [(l.fft_size, l.hop_size, l.win_length) for l in m_list.losses]
    assert got == want

    # and the full Trainer constructs from such a config
    from melgan_amd.engine.trainer import Trainer

    hp.log.log_dir = "/tmp/famres_logs"
    hp.log.chkpt_dir = "/tmp/famres_ck"
    hp.backend.amp = "none"
    hp.backend.fused_adam = False
    tr = Trainer(hp, "famres", enable_tb=False)
    assert len(tr.stft_full.losses) == 3


def test_family_subband_resolutions_key():
    """train.subband_stft_loss_params (family key) overrides the
    sub-band resolution constants; absent -> defaults."""
    from melgan_amd.config import HParam, merge_default, unknown_keys
    from melgan_amd.engine.trainer import Trainer

    raw = """
train:
  subband_stft_loss_params:
    fft_sizes: [256, 512]
    hop_sizes: [20, 40]
    win_lengths: [100, 200]
audio: {hop_length: 256}
model: {out_channels: 4, base_channels: 32, generator_ratio: [8, 4, 2],
        use_subband_stft_loss: true}
log: {log_dir: /tmp/famsub_logs, chkpt_dir: /tmp/famsub_ck}
backend: {amp: none, fused_adam: false}
"""
    hp = merge_default(HParam(raw=raw))
    assert "train.subband_stft_loss_params" not in unknown_keys(hp)
    tr = Trainer(hp, "famsub", enable_tb=False)
    got = [(l.fft_size, l.hop_size, l.win_length) for l in tr.stft_sub.losses]
    assert got == [(256, 20, 100), (512, 40, 200)]
