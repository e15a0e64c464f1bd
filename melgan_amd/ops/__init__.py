from melgan_amd.ops.wnconv import WNConv1d, WNConvTranspose1d  # noqa: F401
from melgan_amd.ops.pqmf import PQMF  # noqa: F401
from melgan_amd.ops.stft_loss import MultiResolutionSTFTLoss  # noqa: F401
