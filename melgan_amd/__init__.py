"""melgan_amd — MI355X-native multi-band MelGAN vocoder framework.

A from-scratch rebuild of the capability set of ``diver-j/melgan-multi``
(reference mount was empty; capability contract reconstructed in
/root/repo/SURVEY.md) designed MI355X-first:

- PyTorch-ROCm as the autograd/orchestration substrate
- hand-written HIP/CDNA4 (gfx950) kernels for the hot ops
  (fused weight-norm conv1d / conv_transpose1d / grouped disc convs,
  PQMF filter bank, STFT-loss epilogues, multi-tensor Adam)
- RCCL over xGMI for data-parallel training (bucketed all-reduce
  overlapped with backward on a dedicated HIP stream)

Layers:
    melgan_amd.config    — YAML config (HParam-compatible schema)
    melgan_amd.models    — Generator / MultiScaleDiscriminator
    melgan_amd.ops       — autograd ops (HIP kernels with PyTorch fallback)
    melgan_amd.data      — datasets (disk mels + synthetic bench mels)
    melgan_amd.engine    — training loop, validation, writer, checkpoint
    melgan_amd.parallel  — native DP engine (bucketed RCCL all-reduce)
"""

__version__ = "0.1.0"

from melgan_amd.config import HParam, load_hparam  # noqa: F401
