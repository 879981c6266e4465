from shifu_amd.train.metrics import TrainingIntermediateResult, EpochStats  # noqa: F401
from shifu_amd.train.trainer import Trainer  # noqa: F401
