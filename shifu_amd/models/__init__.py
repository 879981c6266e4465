from shifu_amd.models.mlp import ShifuMLP, build_model  # noqa: F401
from shifu_amd.models.wide_deep import WideDeep  # noqa: F401
from shifu_amd.models.deepfm import DeepFM  # noqa: F401
