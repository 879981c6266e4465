"""shifu_amd — an MI355X-native distributed tabular-DNN training framework.

A from-scratch rebuild of the capabilities of ShifuML/shifu-tensorflow
(reference: a YARN ApplicationMaster driving a TensorFlow-1.x parameter-server
cluster — see /root/reference) as a single-node 8x AMD Instinct
MI355X framework:

* Python launcher + PyTorch-ROCm tensor shell (one process per GPU),
* hand-written HIP/CDNA4 (gfx950) kernels for the hot ops
  (fused dense GEMM+bias+activation, weighted losses, fused optimizers on a
  flat parameter arena, categorical embedding gather/scatter-add),
* RCCL over xGMI for gradient aggregation (replaces the reference's
  SyncReplicasOptimizer gRPC parameter-server plane,
  reference: shifu-tensorflow-on-yarn/src/main/resources/ssgd_monitor.py:136-142),
* the same public data/config API: ModelConfig.json / ColumnConfig.json and
  gzip'd '|'-delimited normalized CSV with target/weight columns
  (reference: ssgd_monitor.py:348-454),
* the same export contract (GenericModelConfig.json + model artifacts) that
  shifu-tensorflow-eval consumes
  (reference: shifu-tensorflow-eval/.../TensorflowModel.java:111-172).
"""

__version__ = "0.1.0"

from shifu_amd.config.model_config import ModelConfig, ColumnConfig  # noqa: F401
from shifu_amd.config.run_config import RunConfig  # noqa: F401
