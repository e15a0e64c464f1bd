from melgan_amd.engine.trainer import Trainer  # noqa: F401
from melgan_amd.engine.checkpoint import (  # noqa: F401
    save_checkpoint,
    load_checkpoint,
    load_flexible,
)
