"""MI355X-native gradient-accumulation training engine.

A from-scratch rebuild of the capabilities of
``hpandana/gradient-accumulation-tf-estimator`` (TF1/tf.estimator) as a
PyTorch-ROCm + hand-written CDNA4 HIP kernel + RCCL/xGMI framework.
See SURVEY.md at the repo root for the reference layer map this implements.
"""

from .engine.optimizer import TrainOp, create_optimizer
from .engine.accum import AccumEngine
from .engine.schedule import learning_rate
from .engine.flat import DEFAULT_EXCLUDE_FROM_WEIGHT_DECAY

__version__ = "0.1.0"

__all__ = [
    "TrainOp",
    "create_optimizer",
    "AccumEngine",
    "learning_rate",
    "DEFAULT_EXCLUDE_FROM_WEIGHT_DECAY",
]
