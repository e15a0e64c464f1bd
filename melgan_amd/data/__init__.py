from melgan_amd.data.dataset import (  # noqa: F401
    MelFromDisk,
    SyntheticMelDataset,
    create_dataloader,
)
from melgan_amd.data.mel import MelSpectrogram  # noqa: F401
