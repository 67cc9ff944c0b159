"""moco_amd — an MI355X-native momentum-contrast (MoCo v1/v2) pretraining
engine: PyTorch-ROCm autograd substrate, hand-written gfx950 HIP kernels for
the hot ops, RCCL over xGMI for the collectives.

Built from scratch with the capabilities of the reference MoCo codebase
(see SURVEY.md); not a port.
"""

__version__ = "0.2.0"

from . import builder, loader, models, ops, parallel, utils  # noqa: F401
from .builder import MoCo, moco_loss  # noqa: F401
