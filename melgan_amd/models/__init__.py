from melgan_amd.models.generator import Generator, ResStack  # noqa: F401
from melgan_amd.models.discriminator import (  # noqa: F401
    MultiScaleDiscriminator,
    DiscriminatorTower,
)


def build_generator(hp):
    """Construct a Generator from an HParam tree (reference YAML schema)."""
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
