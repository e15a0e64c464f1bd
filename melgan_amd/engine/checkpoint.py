"""Checkpoint save/load in the reference family format.

Contract: torch.save({'model_g', 'model_d', 'optim_g',
'optim_d', 'step', 'hp_str'}, f'{name}_{step:04d}.pt'); resume restores
all of it and warns when hp_str differs; inference loads only model_g.
Generator weights keep the family's weight-norm parameter naming
(*.weight_g / *.weight_v / *.bias) — our fused modules register those
names directly, and `load_flexible` additionally maps checkpoints whose
module PATHS differ (e.g. a reference tree) by ordered shape matching.
"""

from __future__ import annotations

import os

import torch


def save_checkpoint(path_dir: str, name: str, step: int, generator,
                    discriminator, optim_g, optim_d, hp_str: str) -> str:
    os.makedirs(path_dir, exist_ok=True)
    path = os.path.join(path_dir, f"{name}_{step:04d}.pt")
    torch.save({
        "model_g": generator.state_dict(),
        "model_d": discriminator.state_dict(),
        "optim_g": optim_g.state_dict(),
        "optim_d": optim_d.state_dict(),
        "step": step,
        "hp_str": hp_str,
    }, path)
    return path


def load_checkpoint(path: str, map_location="cpu") -> dict:
    return torch.load(path, map_location=map_location, weights_only=False)


def _order_key(name: str):
    """Sort key preserving module order from state-dict insertion order."""
    return 0


def load_flexible(module: torch.nn.Module, state: dict, strict_first=True,
                  logger=None) -> None:
    """Load `state` into `module`; fall back to ordered shape matching.

    The fallback walks both state dicts in insertion order and matches
    parameters by (suffix kind, shape) — this absorbs module-path renames
    between the reference tree (e.g. `generator.generator.0.weight_v`)
    and ours (`layers.0.weight_v`) while refusing shape mismatches.
    """
    try:
        module.load_state_dict(state)
        return
    except (RuntimeError, KeyError):
        if strict_first and logger is not None:
            logger.info("strict checkpoint load failed; trying ordered "
                        "shape-matched load")
    own = module.state_dict()
    own_items = [(k, v) for k, v in own.items()]
    src_items = [(k, v) for k, v in state.items()
                 if isinstance(v, torch.Tensor)]

    def kind(key: str) -> str:
        return key.rsplit(".", 1)[-1]

    mapped = {}
    si = 0
    for ok, ov in own_items:
        found = False
        for j in range(si, len(src_items)):
            sk, sv = src_items[j]
            if kind(sk) == kind(ok) and tuple(sv.shape) == tuple(ov.shape):
                mapped[ok] = sv
                si = j + 1
                found = True
                break
        if not found:
            raise RuntimeError(
                f"checkpoint translation failed: no source tensor for "
                f"{ok} (shape {tuple(ov.shape)}, kind {kind(ok)})")
    module.load_state_dict(mapped)
