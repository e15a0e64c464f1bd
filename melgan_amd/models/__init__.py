from melgan_amd.models.generator import Generator, ResStack  # noqa: F401
from melgan_amd.models.discriminator import (  # noqa: F401
    MultiScaleDiscriminator,
    DiscriminatorTower,
)


def build_generator(hp):
    """Construct a Generator from an HParam tree (reference YAML schema)."""
    prod = 1
    for r in hp.model.generator_ratio:
        prod *= r
    if prod * hp.model.out_channels != hp.audio.hop_length:
        raise ValueError(
            f"model geometry mismatch: prod(generator_ratio)={prod} x "
            f"out_channels={hp.model.out_channels} must equal "
            f"audio.hop_length={hp.audio.hop_length} — the generator has "
            "to upsample mel frames back to waveform samples (otherwise "
            "training fails with a cryptic STFT-loss shape error)")
    return Generator(
        mel_channels=hp.audio.n_mel_channels,
        out_channels=hp.model.out_channels,
        base_channels=hp.model.base_channels,
        ratios=hp.model.generator_ratio,
        n_residual_layers=hp.model.n_residual_layers,
    )


def build_discriminator(hp):
    return MultiScaleDiscriminator(
        num_D=hp.model.num_D,
        ndf=hp.model.ndf,
        n_layers=hp.model.n_layers_D,
        downsamp_factor=hp.model.downsamp_factor,
    )
