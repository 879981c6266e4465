from shifu_amd.config.model_config import ModelConfig, ColumnConfig  # noqa: F401
from shifu_amd.config.run_config import RunConfig  # noqa: F401
