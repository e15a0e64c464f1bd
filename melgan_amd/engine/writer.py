"""Metrics/observability: TensorBoard surface (optional) + JSONL.

Contract: reference logs scalars/audio/figures to
TensorBoard and text logs per run.  We keep that surface when
tensorboard is importable and ALWAYS write a machine-readable JSONL
metrics stream (step, losses, mel-frames/sec, comm bytes) so benchmarks
don't need TB.
"""

from __future__ import annotations

import json
import logging
import os
import time


def get_logger(log_dir: str, name: str) -> logging.Logger:
    os.makedirs(log_dir, exist_ok=True)
    logger = logging.getLogger(f"melgan_amd.{name}")
    logger.setLevel(logging.INFO)
    if not logger.handlers:
        fh = logging.FileHandler(os.path.join(log_dir, f"{name}.log"))
        fh.setFormatter(logging.Formatter(
            "%(asctime)s %(levelname)s %(message)s"))
        logger.addHandler(fh)
        sh = logging.StreamHandler()
        sh.setFormatter(logging.Formatter("%(asctime)s %(message)s"))
        logger.addHandler(sh)
    return logger


class MetricsWriter:
    def __init__(self, log_dir: str, name: str, enable_tb: bool = True,
                 rank: int = 0):
        self.rank = rank
        self.tb = None
        if rank != 0:
            return
        os.makedirs(log_dir, exist_ok=True)
        self.jsonl = open(os.path.join(log_dir, f"{name}_metrics.jsonl"), "a")
        if enable_tb:
            try:
                from torch.utils.tensorboard import SummaryWriter

                self.tb = SummaryWriter(os.path.join(log_dir, name))
            except Exception:
                self.tb = None

    def log_training(self, step: int, **scalars):
        if self.rank != 0:
            return
        rec = {"step": step, "t": time.time(), **{
            k: (float(v) if v is not None else None) for k, v in scalars.items()}}
        self.jsonl.write(json.dumps(rec) + "\n")
        self.jsonl.flush()
        if self.tb is not None:
            for k, v in scalars.items():
                if v is not None:
                    self.tb.add_scalar(f"train/{k}", float(v), step)

    def log_validation(self, step: int, audio=None, sr: int = 22050, **scalars):
        if self.rank != 0:
            return
        rec = {"step": step, "t": time.time(), "phase": "val", **{
            k: (float(v) if v is not None else None) for k, v in scalars.items()}}
        self.jsonl.write(json.dumps(rec) + "\n")
        self.jsonl.flush()
        if self.tb is not None:
            for k, v in scalars.items():
                if v is not None:
                    self.tb.add_scalar(f"val/{k}", float(v), step)
            if audio is not None:
                try:
                    self.tb.add_audio("val/audio", audio.reshape(1, -1),
                                      step, sample_rate=sr)
                    fig = _spec_figure(audio)
                    if fig is not None:
                        self.tb.add_figure("val/spectrogram", fig, step)
                except Exception:
                    pass

    def close(self):
        if self.rank == 0:
            self.jsonl.close()
            if self.tb is not None:
                self.tb.close()


def _spec_figure(audio, n_fft: int = 1024, hop: int = 256):
    """Matplotlib spectrogram figure for TensorBoard (family parity)."""
    try:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        import torch

        spec = torch.stft(audio.reshape(1, -1).float(), n_fft, hop,
                          window=torch.hann_window(n_fft),
                          return_complex=True).abs()[0]
        fig, ax = plt.subplots(figsize=(8, 3))
        ax.imshow(spec.clamp_min(1e-5).log().numpy(), origin="lower",
                  aspect="auto", cmap="magma")
        ax.set_xlabel("frame")
        ax.set_ylabel("bin")
        fig.tight_layout()
        return fig
    except Exception:
        return None
