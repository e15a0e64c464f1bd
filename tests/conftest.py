import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an MI355X GPU (run with -m gpu on a GPU box)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tiny_hp():
    """Small single-band config for fast CPU tests."""
    from melgan_amd.config import HParam, merge_default

    raw = """
data: {train_dir: null, val_dir: null}
train:
  batch_size: 2
  num_workers: 0
  discriminator_train_start_steps: 0
audio:
  n_mel_channels: 80
  segment_length: 4096
  hop_length: 256
  filter_length: 1024
  win_length: 1024
  sampling_rate: 22050
model:
  out_channels: 1
  base_channels: 64
  generator_ratio: [8, 8, 2, 2]
  n_residual_layers: 3
log: {summary_interval: 2, validation_interval: 100, save_interval: 100,
      chkpt_dir: /tmp/melgan_test_chkpt, log_dir: /tmp/melgan_test_logs}
backend: {amp: none, fused_adam: false}
"""
    return merge_default(HParam(raw=raw))


@pytest.fixture
def tiny_mb_hp():
    """Small 4-band config for fast CPU tests."""
    from melgan_amd.config import HParam, merge_default

    raw = """
data: {train_dir: null, val_dir: null}
train:
  batch_size: 2
  num_workers: 0
  discriminator_train_start_steps: 1
audio:
  n_mel_channels: 80
  segment_length: 4096
  hop_length: 256
model:
  out_channels: 4
  base_channels: 64
  generator_ratio: [8, 4, 2]
  n_residual_layers: 3
  use_subband_stft_loss: true
log: {summary_interval: 2, validation_interval: 100, save_interval: 100,
      chkpt_dir: /tmp/melgan_test_chkpt_mb, log_dir: /tmp/melgan_test_logs_mb}
backend: {amp: none, fused_adam: false}
"""
    return merge_default(HParam(raw=raw))
