"""RunConfig — the single launcher config that replaces the reference's
three-layer Hadoop-XML cascade (global-default.xml -> global.xml ->
global-final.xml, reference: TensorflowClient.java:212-224,389-403) and the
Java->Python env-var contract.

One JSON (or CLI flags) drives the whole single-node run: worker count = GPUs,
data paths, model paths, epochs, batch size, all-reduce bucket size.
"""
from __future__ import annotations

import dataclasses
import json
import os
from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class RunConfig:
    # -- topology --
    num_gpus: int = 1                    # successor of shifu.worker.instances
    backend: str = "auto"                # "nccl" (RCCL on ROCm) | "gloo" | "auto"
    master_addr: str = "127.0.0.1"
    master_port: int = 29511

    # -- data (successor of TRAINING_DATA_PATH / TOTAL_TRAINING_DATA_NUMBER) --
    training_data_path: List[str] = field(default_factory=list)  # csv(.gz) files
    delimiter: str = "|"
    target_column: int = 0
    weight_column: int = -1
    selected_numeric_columns: List[int] = field(default_factory=list)
    selected_categorical_columns: List[int] = field(default_factory=list)
    vocab_sizes: List[int] = field(default_factory=list)  # per categorical column
    valid_set_rate: float = 0.2
    seed: int = 1234

    # -- model family --
    model_type: str = "auto"   # "mlp" | "wide_deep" | "deepfm" | "auto"
    embed_dim: int = 16
    unified_arena: bool = True # wide scalar weights live as column D of the
                               # deep [R, D+2] arena: one gather/scatter/EP
                               # exchange for both (models/wide_deep.py);
                               # False keeps two separate arenas
    emb_mode: str = "auto"     # "ep"/"ep_table": feature-sharded arenas with
                               # STATIC all-to-all splits (parallel/ep.py
                               # TableShardedEmbedding — no per-step host
                               # sync, graph-capturable); "ep_row": row%world
                               # sharding (single table > 1 GPU's HBM);
                               # "dp": replicated + sparse-allgather grads;
                               # "auto" = ep at world>1 (the xGMI-native
                               # choice), dp when quorum_ratio < 1

    # -- model / training --
    model_config_path: Optional[str] = None
    column_config_path: Optional[str] = None
    tmp_model_path: str = "./tmp_model"      # checkpoint dir (TMP_MODEL_PATH)
    final_model_path: str = "./final_model"  # export dir (FINAL_MODEL_PATH)
    epochs: Optional[int] = None             # overrides ModelConfig numTrainEpochs
    batch_size: Optional[int] = None         # per-rank batch size override
    dtype: str = "bf16"                      # compute dtype on GPU ("bf16"|"fp32")
    data_residency: str = "auto"             # "auto"/"device": whole shard in
                                             # HBM (fastest); "stream": shard
                                             # stays in pinned host RAM, only
                                             # batches cross PCIe (shards
                                             # beyond-HBM datasets)

    # -- distributed knobs (xGMI-tuned) --
    bucket_mb: int = 128       # all-reduce bucket size; ring all-reduce over
                               # 7x153GB/s p2p links is per-link bound -> big buckets
    overlap_allreduce: bool = True
    quorum_ratio: float = 1.0  # REPLICAS_TO_AGGREGATE_RATIO analog (ssgd.py:19;
                               # SAGN.py:161 uses 0.9): q < 1 aggregates each
                               # sync step from only ceil(q*world) ranks on a
                               # deterministic rotation (parallel/dist.py
                               # GradAggregator); requires emb_mode="dp".

    # -- robustness (successor of heartbeat/backup machinery) --
    heartbeat_interval_s: float = 1.0    # shifu.task.heartbeat-interval default 1000ms
    max_missed_heartbeats: int = 25      # shifu.task.max-missed-heartbeats default 25
    max_rank_restarts: int = 1           # restart a dead rank from last checkpoint
    startup_grace_s: float = 360.0       # heartbeat budget before a rank's FIRST
                                         # message: model/arena init can take
                                         # minutes (successor of the reference's
                                         # 6-min cluster-registration cutover,
                                         # util/Constants.java:92-94)
    checkpoint_every_epochs: int = 1
    checkpoint_every_secs: float = 0.0   # >0: also time-based mid-epoch saves
                                         # (reference: Supervisor save_model_secs,
                                         #  ssgd.py:124-128); rank-local clocks,
                                         # so disabled for EP-sharded arenas
    checkpoint_every_steps: int = 0      # >0: mid-epoch saves every N sync
                                         # steps — rank-synchronized (uniform
                                         # stepping), so valid for EP shards
                                         # too

    # -- misc --
    data_cache: bool = True              # cache parsed shards under
                                         # log_dir/shard_cache so a launcher
                                         # restart skips the CSV re-parse
                                         # (keyed on file sizes/mtimes +
                                         # column spec)
    log_dir: str = "./logs"
    device: str = "auto"                 # "cuda" | "cpu" | "auto"
    enable_trace: bool = False           # per-phase step timing (utils/trace.py)
    graphs: str = "auto"                 # hipGraph-captured steps: "auto"|"on"|"off"
                                         # (auto = on for single-GPU cuda, full
                                         # batches, no window mode)

    def resolved_backend(self) -> str:
        if self.backend != "auto":
            return self.backend
        import torch
        return "nccl" if torch.cuda.is_available() else "gloo"

    def resolved_device(self) -> str:
        if self.device != "auto":
            return self.device
        import torch
        return "cuda" if torch.cuda.is_available() else "cpu"

    @classmethod
    def load(cls, path: str) -> "RunConfig":
        with open(path, "r") as f:
            d = json.load(f)
        known = {f.name for f in dataclasses.fields(cls)}
        return cls(**{k: v for k, v in d.items() if k in known})

    def save(self, path: str) -> None:
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        with open(path, "w") as f:
            json.dump(dataclasses.asdict(self), f, indent=2)

    def apply_column_config(self, cc) -> None:
        """Derive column selections from a parsed ColumnConfig (replaces the
        pre-digested SELECTED_*_COLUMN_NUMS env vars)."""
        self.target_column = cc.target_column
        self.weight_column = cc.weight_column
        self.selected_numeric_columns = cc.selected_numeric_columns
        self.selected_categorical_columns = cc.selected_categorical_columns
        vs = cc.vocab_sizes()
        self.vocab_sizes = [vs[c] for c in self.selected_categorical_columns]

    def resolved_model_type(self) -> str:
        if self.model_type != "auto":
            return self.model_type
        return "wide_deep" if self.selected_categorical_columns else "mlp"
