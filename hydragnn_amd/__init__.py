"""hydragnn_amd — MI355X-native multi-headed GNN training framework.

Brand-new CDNA4-first implementation of the ORNL/HydraGNN capability
set: PyTorch-ROCm + hand-written HIP/gfx950 kernels for
the message-passing hot path + RCCL over xGMI for data/model
parallelism.  The public API mirrors the reference
(/root/reference/hydragnn/__init__.py:11): preprocess, models, train,
postprocess, utils.
"""

from . import data, models, ops, postprocess, preprocess, train, utils  # noqa
from .data import Batch, Data
from .run import run_training, run_prediction

__version__ = "0.1.0"
