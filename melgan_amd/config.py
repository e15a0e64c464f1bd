"""YAML config system, schema-compatible with the reference family.

The reference (`diver-j/melgan-multi`, SURVEY.md §2.1 "Config system",
expected path utils/hparams.py [R]) exposes an attribute-access wrapper
over a YAML file with groups ``data: train: audio: model: log:`` and keeps
the raw YAML string (``hp_str``) which is stored inside checkpoints and
compared on resume.  This module reproduces that contract and adds an
additive ``backend:`` group (ours) for MI355X kernel/comm tuning knobs —
reference configs load unchanged.
"""

from __future__ import annotations

import os

import yaml


class Dotdict(dict):
    """dict with attribute access, recursively applied (family convention).

    Missing keys return ``None`` (the family's Dotdict behaviour) so that
    optional config groups can be probed with ``hp.model.foo is None``.
    """

    def __getattr__(self, name):
        if name.startswith("__"):  # keep pickling / copy sane
            raise AttributeError(name)
        return self.get(name)

    def __setattr__(self, name, value):
        self[name] = value

    def __delattr__(self, name):
        try:
            del self[name]
        except KeyError:
            raise AttributeError(name)


def _wrap(obj):
    if isinstance(obj, dict):
        return Dotdict({k: _wrap(v) for k, v in obj.items()})
    if isinstance(obj, (list, tuple)):
        return type(obj)(_wrap(v) for v in obj)
    return obj


def load_hparam_str(hp_str: str) -> "HParam":
    """Build an HParam from a raw YAML string (used on checkpoint resume)."""
    return HParam(raw=hp_str)


def load_hparam(path: str) -> "HParam":
    return HParam(path)


class HParam(Dotdict):
    """Attribute-access hyperparameter tree loaded from YAML.

    ``HParam(path)`` loads a file; ``HParam(raw=yaml_string)`` parses a
    string (checkpoint ``hp_str`` round-trip).  The raw text is kept in
    ``hp_str`` exactly as read, so checkpoints can embed and compare it.
    """

    def __init__(self, path: str | None = None, raw: str | None = None):
        if (path is None) == (raw is None):
            raise ValueError("HParam needs exactly one of path= or raw=")
        if path is not None:
            with open(path, "r") as f:
                raw = f.read()
        data = yaml.safe_load(raw) or {}
        if not isinstance(data, dict):
            raise ValueError("config root must be a YAML mapping")
        super().__init__(_wrap(data))
        # stash raw string without making it a config key
        object.__setattr__(self, "_raw", raw)

    def __getattr__(self, name):
        if name == "hp_str":
            return object.__getattribute__(self, "_raw")
        return super().__getattr__(name)


def merge_default(hp: HParam) -> HParam:
    """Fill defaults for keys the reference family treats as optional."""
    # Groups
    for g in ("data", "train", "audio", "model", "log", "backend"):
        if hp.get(g) is None:
            hp[g] = Dotdict()
    hp.audio.setdefault("n_mel_channels", 80)
    hp.audio.setdefault("sampling_rate", 22050)
    hp.audio.setdefault("hop_length", 256)
    hp.audio.setdefault("filter_length", 1024)
    hp.audio.setdefault("win_length", 1024)
    hp.audio.setdefault("segment_length", 16000)
    hp.audio.setdefault("pad_short", 2000)
    hp.audio.setdefault("mel_fmin", 0.0)
    hp.audio.setdefault("mel_fmax", 8000.0)
    hp.model.setdefault("out_channels", 1)
    hp.model.setdefault("base_channels", 512)
    hp.model.setdefault("n_residual_layers", 3)
    hp.model.setdefault("num_D", 3)
    hp.model.setdefault("ndf", 16)
    hp.model.setdefault("n_layers_D", 4)
    hp.model.setdefault("downsamp_factor", 4)
    hp.model.setdefault("feat_match", 10.0)
    hp.model.setdefault("use_subband_stft_loss", hp.model.out_channels > 1)
    if hp.model.get("generator_ratio") is None:
        # total upsampling = hop_length / out_channels
        total = hp.audio.hop_length // hp.model.out_channels
        hp.model.generator_ratio = _default_ratios(total)
    hp.train.setdefault("batch_size", 16)
    hp.train.setdefault("num_workers", 2)
    hp.train.setdefault("seed", 1234)
    hp.train.setdefault("max_steps", 10 ** 9)
    hp.train.setdefault("discriminator_train_start_steps", 200000)
    hp.train.setdefault("adam", Dotdict())
    hp.train.adam.setdefault("lr", 1e-4)
    hp.train.adam.setdefault("beta1", 0.5)
    hp.train.adam.setdefault("beta2", 0.9)
    hp.train.setdefault("stft_loss_params", None)
    hp.log.setdefault("summary_interval", 20)
    hp.log.setdefault("validation_interval", 1000)
    hp.log.setdefault("save_interval", 1000)
    hp.log.setdefault("chkpt_dir", "chkpt")
    hp.log.setdefault("log_dir", "logs")
    # MI355X backend knobs (additive; absent in reference configs)
    hp.backend.setdefault("use_hip_kernels", True)
    hp.backend.setdefault("amp", "bf16")
    hp.backend.setdefault("bucket_cap_mb", 25)
    hp.backend.setdefault("fused_adam", True)
    return hp


def _default_ratios(total: int):
    """Factor a total upsampling ratio into per-block ratios, largest first.

    256 -> [8, 8, 2, 2]; 64 -> [8, 4, 2]; 32 -> [8, 4]; 300 -> [5,5,4,3].
    """
    known = {
        256: [8, 8, 2, 2], 128: [8, 8, 2], 64: [8, 4, 2], 32: [8, 4],
        512: [8, 8, 8], 300: [5, 5, 4, 3], 75: [5, 5, 3], 120: [5, 4, 3, 2],
    }
    if total in known:
        return known[total]
    ratios = []
    rem = total
    for r in (8, 7, 6, 5, 4, 3, 2):
        while rem % r == 0 and rem > 1:
            ratios.append(r)
            rem //= r
    if rem != 1:
        ratios.append(rem)
    return ratios


# keys the framework reads (top two levels); used for typo warnings only —
# unknown keys are reported, never rejected (reference configs may carry
# extra family keys)
_KNOWN_KEYS = {
    "data": {"train_dir", "val_dir"},
    "train": {"batch_size", "num_workers", "seed", "max_steps",
              "discriminator_train_start_steps", "adam", "stft_loss_params",
              "subband_stft_loss_params",
              "lambda_adv"},
    "audio": {"n_mel_channels", "sampling_rate", "hop_length",
              "filter_length", "win_length", "segment_length", "pad_short",
              "mel_fmin", "mel_fmax"},
    "model": {"out_channels", "base_channels", "n_residual_layers",
              "generator_ratio", "num_D", "ndf", "n_layers_D",
              "downsamp_factor", "feat_match", "use_subband_stft_loss",
              "pqmf_cutoff"},
    "log": {"summary_interval", "validation_interval", "save_interval",
            "chkpt_dir", "log_dir"},
    "backend": {"use_hip_kernels", "amp", "bucket_cap_mb", "fused_adam",
                "loss_stream", "d_stream", "debug_sync", "phase_timing"},
}


def unknown_keys(hp) -> list[str]:
    """Dotted paths of config keys the framework never reads (typo check)."""
    out = []
    for g, sub in hp.items():
        if g not in _KNOWN_KEYS:
            out.append(g)
        elif isinstance(sub, dict):
            out.extend(f"{g}.{k}" for k in sub if k not in _KNOWN_KEYS[g])
    return out


def check_hp_compat(old_hp_str: str, new_hp_str: str, logger=None) -> bool:
    """Family behaviour on resume: warn (not fail) when hp_str differs."""
    same = old_hp_str == new_hp_str
    if not same and logger is not None:
        logger.warning(
            "New hparams are different from the checkpoint's. "
            "Will use new hparams."
        )
    return same


def env_world() -> tuple[int, int, int]:
    """(rank, local_rank, world_size) from torchrun env, defaulting to 1p."""
    return (
        int(os.environ.get("RANK", "0")),
        int(os.environ.get("LOCAL_RANK", "0")),
        int(os.environ.get("WORLD_SIZE", "1")),
    )
