"""Integration smoke over every SHIPPED config ("training
steps of each of the driver configs on synthetic mels, losses finite") —
the tiny-fixture smokes in test_train_smoke.py don't exercise the real
YAMLs' geometry (base 512 channels, per-config ratios/hops/bands)."""

import glob
import os

import pytest
import torch

from melgan_amd.config import HParam, merge_default
from melgan_amd.engine.trainer import Trainer

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CONFIGS = sorted(glob.glob(os.path.join(REPO, "configs", "*.yaml")))


@pytest.mark.timeout(600)
@pytest.mark.parametrize("cfg", CONFIGS, ids=os.path.basename)
def test_two_full_gan_steps_each_shipped_config(cfg, tmp_path):
    hp = merge_default(HParam(cfg))
    # shrink to CPU scale without touching the model geometry
    hp.train.batch_size = 1
    hp.train.num_workers = 0
    hp.audio.segment_length = 16 * hp.audio.hop_length
    hp.train.discriminator_train_start_steps = 0
    hp.backend.amp = "none"
    hp.backend.fused_adam = False
    hp.log.chkpt_dir = str(tmp_path / "ck")
    hp.log.log_dir = str(tmp_path / "lg")

    torch.manual_seed(0)
    tr = Trainer(hp, "cfgsmoke", device=torch.device("cpu"),
                 enable_tb=False)
    frames = hp.audio.segment_length // hp.audio.hop_length
    mel = torch.randn(1, hp.audio.n_mel_channels, frames) * 2 - 6
    audio = torch.rand(1, 1, hp.audio.segment_length) * 1.6 - 0.8
    for _ in range(2):
        g_loss, parts, d_loss = tr.train_step(mel, audio)
        assert torch.isfinite(g_loss), cfg
        assert d_loss is not None and torch.isfinite(d_loss), cfg
    assert parts["adv"] is not None and parts["feat"] is not None
