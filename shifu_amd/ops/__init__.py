from shifu_amd.ops.dispatch import hip_available, hip_ops, require_hip  # noqa: F401
from shifu_amd.ops.linear import FusedLinear, fused_linear  # noqa: F401
from shifu_amd.ops.loss import weighted_loss  # noqa: F401
from shifu_amd.ops.embedding import MultiEmbedding  # noqa: F401
from shifu_amd.ops.flat import FlatParams  # noqa: F401
from shifu_amd.ops.optim import FusedOptimizer  # noqa: F401
from shifu_amd.ops.fm import fm_second_order  # noqa: F401
