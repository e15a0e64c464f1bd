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
    # the saved dict is plain tensors/ints/str, so the safe loader works;
    # fall back only for legacy pickles (e.g. foreign optimizer states)
    try:
        return torch.load(path, map_location=map_location, weights_only=True)
    except Exception:
        return torch.load(path, map_location=map_location,
                          weights_only=False)


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

    def kind(key: str) -> str:
        return key.rsplit(".", 1)[-1]

    def prefix(key: str) -> str:
        return key.rsplit(".", 1)[0] if "." in key else ""

    def modules_in_order(items):
        """[(prefix, {kind: (key, tensor)})] preserving insertion order.
        A module = the group of tensors sharing a key prefix (one conv's
        weight_v/weight_g/bias, in whatever per-module key ORDER the
        source library emits them)."""
        groups = []
        index = {}
        for k, v in items:
            p = prefix(k)
            if p not in index:
                index[p] = len(groups)
                groups.append((p, {}))
            groups[index[p]][1][kind(k)] = (k, v)
        return groups

    src_groups = modules_in_order(
        [(k, v) for k, v in state.items() if isinstance(v, torch.Tensor)])
    own_groups = modules_in_order(list(own.items()))

    mapped = {}
    si = 0
    for op, okinds in own_groups:
        found = False
        for j in range(si, len(src_groups)):
            sp, skinds = src_groups[j]
            if set(okinds) == set(skinds) and all(
                    tuple(skinds[t][1].shape) == tuple(okinds[t][1].shape)
                    for t in okinds):
                for t in okinds:
                    mapped[okinds[t][0]] = skinds[t][1]
                si = j + 1
                found = True
                break
        if not found:
            shapes = {t: tuple(v.shape) for t, (_, v) in okinds.items()}
            raise RuntimeError(
                f"checkpoint translation failed: no source module for "
                f"'{op}' with tensors {shapes}")
    module.load_state_dict(mapped)
