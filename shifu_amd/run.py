"""Default end-to-end run path + CLI.

`python -m shifu_amd.run --run-config run.json [--model-config ModelConfig.json]`
is the successor of the reference's `shifu train` submission
(client/TensorflowClient.java:290): it loads configs, launches one rank per
GPU, trains, and exports — all on the local node.
"""
from __future__ import annotations

import argparse
import os
from typing import Optional

import torch

from shifu_amd.config.model_config import ColumnConfig, ModelConfig
from shifu_amd.config.run_config import RunConfig
from shifu_amd.data.csv_loader import list_training_files, load_csv_files
from shifu_amd.data.sharding import shard_files, shard_rows
from shifu_amd.models.mlp import build_model
from shifu_amd.parallel.dist import destroy_distributed, init_distributed
from shifu_amd.parallel.launcher import Launcher
from shifu_amd.train.trainer import Trainer


def default_rank_entry(rank: int, world: int, rc: RunConfig, mc: ModelConfig,
                       metric_sink, heartbeat) -> None:
    """One rank: init dist, load+shard data, build model, fit."""
    init_distributed(rc.resolved_backend(), rc.master_addr, rc.master_port)
    try:
        files = list_training_files(rc.training_data_path)
        from shifu_amd.io import load_csv_native
        from shifu_amd.data.shard_cache import load_split_cached
        import numpy as np

        spec = {"num": rc.selected_numeric_columns,
                "cat": rc.selected_categorical_columns,
                "target": rc.target_column, "weight": rc.weight_column,
                "delim": rc.delimiter, "valid": rc.valid_set_rate,
                "seed": rc.seed}
        cache_dir = (os.path.join(rc.log_dir, "shard_cache")
                     if getattr(rc, "data_cache", True) else None)

        def build():
            if world > 1 and len(files) >= world:
                # reference behavior (TrainingDataSet.java:55-89): round-robin
                # FILES across workers, each rank parses only its shard — the
                # 100M-row config never parses the full dataset per rank.
                # Valid split is then per-rank (seeded per rank, like the
                # reference's per-worker random split).
                my_files = shard_files(files, rank, world)
                shard = load_csv_native(my_files, rc.selected_numeric_columns,
                                        rc.selected_categorical_columns,
                                        rc.target_column, rc.weight_column,
                                        rc.delimiter)
                return shard.split(rc.valid_set_rate, seed=rc.seed + rank)
            # few big files: every rank parses, then row-range shards; a
            # shared-seed global split keeps valid metrics identical across
            # ranks
            full = load_csv_native(files, rc.selected_numeric_columns,
                                   rc.selected_categorical_columns,
                                   rc.target_column, rc.weight_column,
                                   rc.delimiter)
            tr_, va_ = full.split(rc.valid_set_rate, seed=rc.seed)
            s, e = shard_rows(len(tr_), rank, world)
            return tr_.subset(np.arange(s, e)), va_

        # restart-from-checkpoint skips the CSV re-parse: the parsed shard is
        # cached keyed on file sizes/mtimes + column spec (data/shard_cache.py)
        train, valid, cached = load_split_cached(cache_dir, files, spec,
                                                 rank, world, build)

        # per-worker pos/neg counts like the reference's load_data logging
        # (ssgd_monitor.py:448-452)
        print(f"[rank {rank}] train rows={len(train)} pos={train.pos_count} "
              f"neg={train.neg_count}; valid rows={len(valid)}"
              f"{' (shard cache)' if cached else ''}", flush=True)

        device = torch.device(rc.resolved_device(),
                              rank % max(torch.cuda.device_count(), 1)
                              if rc.resolved_device() == "cuda" else 0) \
            if rc.resolved_device() == "cuda" else torch.device("cpu")

        vocab = rc.vocab_sizes or [1000] * len(rc.selected_categorical_columns)
        emb_mode = getattr(rc, "emb_mode", "auto")
        if emb_mode == "auto":
            # table-wise EP (static all-to-all splits) is the xGMI-native
            # default; quorum < 1 needs replicated arenas (dp)
            emb_mode = ("ep" if world > 1 and rc.quorum_ratio >= 1.0 else "dp")
        sharded = {"ep": "table", "ep_table": "table", "ep_row": "row",
                   "dp": False}[emb_mode]
        model = build_model(mc, len(rc.selected_numeric_columns), vocab,
                            model_type=rc.resolved_model_type(),
                            embed_dim=rc.embed_dim, seed=rc.seed,
                            sharded_embeddings=sharded, world=world, rank=rank,
                            unified=getattr(rc, "unified_arena", True))
        trainer = Trainer(model, mc, rc, train, valid, rank=rank,
                          world_size=world, device=device, metric_sink=metric_sink,
                          heartbeat=heartbeat)
        trainer.fit()
    finally:
        destroy_distributed()


def main(argv: Optional[list] = None) -> int:
    ap = argparse.ArgumentParser("shifu_amd.run")
    ap.add_argument("--run-config", required=True)
    ap.add_argument("--model-config", default=None)
    ap.add_argument("--column-config", default=None)
    args = ap.parse_args(argv)

    rc = RunConfig.load(args.run_config)
    mc = ModelConfig.load(args.model_config or rc.model_config_path) \
        if (args.model_config or rc.model_config_path) else ModelConfig()
    cc_path = args.column_config or rc.column_config_path
    if cc_path and os.path.exists(cc_path):
        rc.apply_column_config(ColumnConfig.load(cc_path))

    launcher = Launcher(rc, mc, default_rank_entry)
    stats = launcher.run()
    print(f"finished {len(stats)} epochs; final valid_err="
          f"{stats[-1].mean_valid_error if stats else float('nan'):.6f}")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
