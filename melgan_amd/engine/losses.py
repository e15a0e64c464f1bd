"""GAN loss terms (LSGAN + feature matching + multi-resolution STFT).

Contract (SURVEY.md §2.1 "Training engine"): generator-only warmup with
STFT loss until `discriminator_train_start_steps`; then LSGAN MSE
adversarial losses + feature matching (lambda ~10) + multi-res STFT
(full-band on PQMF-synthesised audio, sub-band on raw generator output).
All loss math in fp32 (bf16 GAN numerics — SURVEY.md §7 hard part 3).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F


def adversarial_g_loss(disc_fake) -> torch.Tensor:
    """LSGAN generator loss: mean over scales of MSE(score, 1)."""
    loss = 0.0
    for feats in disc_fake:
        score = feats[-1].float()
        loss = loss + F.mse_loss(score, torch.ones_like(score))
    return loss / len(disc_fake)


def feature_matching_loss(disc_fake, disc_real) -> torch.Tensor:
    """L1 over all intermediate feature maps (scores excluded).
    On GPU: one multi-tensor kernel each way (K11b) instead of ~60
    elementwise/reduce launches."""
    flat_f, flat_r = [], []
    for feats_f, feats_r in zip(disc_fake, disc_real):
        for f, r in zip(feats_f[:-1], feats_r[:-1]):
            flat_f.append(f)
            flat_r.append(r.detach())
    if not flat_f:
        return torch.zeros(())
    from melgan_amd.ops import _ext

    if _ext.use_hip(flat_f[0]) and _ext.ext() is not None and \
            hasattr(_ext.ext(), "multi_l1_fwd") and \
            all(t.dtype == flat_f[0].dtype for t in flat_f + flat_r):
        from melgan_amd.ops.fused import fused_feature_matching

        return fused_feature_matching(flat_f, flat_r)
    loss = 0.0
    for f, r in zip(flat_f, flat_r):
        loss = loss + F.l1_loss(f.float(), r.float())
    return loss / len(flat_f)


def discriminator_loss(disc_real, disc_fake) -> torch.Tensor:
    """LSGAN: mean over scales of MSE(D(real),1) + MSE(D(fake),0)."""
    loss = 0.0
    for feats_r, feats_f in zip(disc_real, disc_fake):
        sr = feats_r[-1].float()
        sf = feats_f[-1].float()
        loss = loss + F.mse_loss(sr, torch.ones_like(sr)) \
            + F.mse_loss(sf, torch.zeros_like(sf))
    return loss / len(disc_real)
