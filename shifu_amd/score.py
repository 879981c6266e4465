"""Offline batch scoring CLI.

Successor of the reference eval side's dataset scoring (the Java `Computable`
driven over an eval set — shifu-tensorflow-eval/.../TensorflowModel.java:52-94):
read normalized csv(.gz) rows, score them through an exported model, write one
score per row, and optionally report AUC against the target column.

    python -m shifu_amd.score --model final_model/ \
        --data part-*.csv.gz --column-config ColumnConfig.json \
        --output scores.csv [--auc] [--batch 65536] [--device cpu|cuda]

Without --column-config the row layout is assumed to be the exported feature
order (num_dense floats, then categorical ids) with no target/weight columns.

Distributed eval (the reference runs scoring as a parallel Hadoop job):
launch the same command under torchrun — files are round-robined across
ranks (one process per GPU with --device cuda), each rank writes
`<output>.partR`, and rank 0 reports rows + whole-set AUC:

    torchrun --nproc-per-node 8 --master-addr 127.0.0.1 -m shifu_amd.score \
        --model final_model/ --data eval/ --column-config CC.json \
        --output scores.csv --auc --device cuda
"""
from __future__ import annotations

import argparse
import json
import os
import sys
from typing import List, Optional

import numpy as np


def score_files(model_dir: str, data_paths: List[str],
                column_config: Optional[str] = None, batch: int = 65536,
                device: str = "cpu", delimiter: str = "|"):
    """Returns (scores np[N], targets np[N] or None)."""
    import torch
    from shifu_amd.train.export import load_exported
    from shifu_amd.io import load_csv_native

    model = load_exported(model_dir, device=device)
    with open(f"{model_dir}/graph.json") as f:
        spec = json.load(f)
    nd = int(spec["num_dense"])
    nc = len(spec.get("vocab_sizes", []))

    if column_config:
        from shifu_amd.config.model_config import ColumnConfig
        cc = ColumnConfig.load(column_config)
        num_cols = cc.selected_numeric_columns
        cat_cols = cc.selected_categorical_columns
        tgt, wcol = cc.target_column, cc.weight_column
        if len(num_cols) != nd or len(cat_cols) != nc:
            raise ValueError(
                f"ColumnConfig selects {len(num_cols)} numeric + {len(cat_cols)} "
                f"categorical columns but the model expects {nd} + {nc}")
        ds = load_csv_native(data_paths, num_cols, cat_cols, tgt, wcol, delimiter)
        targets = ds.target
    else:
        # raw layout: nd floats then nc ids, no target/weight column at all
        # (target_column=-1: the loader fills targets with zeros, which we
        # discard — column 0 stays a plain dense feature)
        ds = load_csv_native(data_paths, list(range(nd)),
                             list(range(nd, nd + nc)),
                             target_column=-1, weight_column=-1,
                             delimiter=delimiter)
        targets = None

    scores = np.empty(len(ds), dtype=np.float64)
    with torch.no_grad():
        for s in range(0, len(ds), batch):
            e = min(s + batch, len(ds))
            dense = torch.from_numpy(ds.dense[s:e]).to(device)
            if nc:
                cats = torch.from_numpy(ds.cats[s:e]).to(device)
                p = model.predict(dense, cats)
            else:
                p = model.predict(dense)
            scores[s:e] = p.cpu().numpy()
    return scores, targets


def main(argv=None) -> int:
    ap = argparse.ArgumentParser("shifu_amd.score", description=__doc__)
    ap.add_argument("--model", required=True, help="export directory")
    ap.add_argument("--data", required=True, nargs="+", help="csv(.gz) files/dirs")
    ap.add_argument("--column-config", default=None)
    ap.add_argument("--output", default="-", help="scores file ('-' = stdout); "
                    "multi-rank runs write <output>.partR per rank")
    ap.add_argument("--batch", type=int, default=65536)
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--delimiter", default="|")
    ap.add_argument("--auc", action="store_true",
                    help="also print AUC vs the target column (needs --column-config)")
    args = ap.parse_args(argv)

    # Distributed eval (successor of the reference scoring an eval set as a
    # parallel Hadoop job): launch via torchrun — files are round-robined
    # across ranks, each rank scores its shard (one process per GPU with
    # --device cuda), scores land in per-rank part files in file-shard
    # order, and rank 0 reports the AUC over the gathered whole set.
    rank, world = 0, 1
    dist = None
    if os.environ.get("RANK") is not None and os.environ.get("WORLD_SIZE"):
        import torch.distributed as dist
        dist.init_process_group("gloo" if args.device == "cpu" else "nccl")
        rank, world = dist.get_rank(), dist.get_world_size()

    from shifu_amd.data.csv_loader import list_training_files
    from shifu_amd.data.sharding import shard_files
    paths = list_training_files(list(args.data))
    my_paths = shard_files(paths, rank, world) if world > 1 else paths
    if world > 1 and args.device == "cuda":
        import torch
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    scores, targets = (score_files(args.model, my_paths, args.column_config,
                                   args.batch, args.device, args.delimiter)
                       if my_paths else (np.empty(0), None))

    if args.output == "-" and world == 1:
        out = sys.stdout
    else:
        path = args.output if world == 1 else \
            (f"{args.output}.part{rank}" if args.output != "-"
             else f"scores.part{rank}")
        out = open(path, "w")
    try:
        for v in scores:
            out.write(f"{v:.6f}\n")
    finally:
        if out is not sys.stdout:
            out.close()

    if world > 1:
        # gather (score, target) arrays so rank 0 computes the WHOLE-set AUC
        gathered = [None] * world if rank == 0 else None
        dist.gather_object((scores, targets), gathered, dst=0)
        if rank != 0:
            dist.destroy_process_group()
            return 0
        scores = np.concatenate([g[0] for g in gathered])
        tl = [g[1] for g in gathered if g[1] is not None and len(g[1])]
        targets = np.concatenate(tl) if tl else None
        dist.destroy_process_group()

    if world == 1 and dist is not None and dist.is_initialized():
        dist.destroy_process_group()   # torchrun with a single rank

    summary = {"rows": int(len(scores)), "ranks": world}
    if args.auc and targets is not None and len(scores):
        from shifu_amd.train.trainer import auc_score
        summary["auc"] = auc_score(scores, targets)
    print(json.dumps(summary), file=sys.stderr, flush=True)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
