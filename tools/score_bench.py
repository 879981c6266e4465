#!/usr/bin/env python3
"""Serving/scoring throughput bench — the eval-side counterpart of bench.py.

The reference scores eval sets through the Java `Computable` one row at a
time (shifu-tensorflow-eval/.../TensorflowModel.java:52-94).  This measures
our ShifuScorer on an exported headline Wide&Deep bundle:

  * compute()        — per-row latency (the Java call pattern)
  * compute_batch()  — batched scoring rows/s (CPU and, with --device cuda,
                       the GPU path the batch scorer `shifu_amd.score` uses)

Synthetic model + data (no network).  One JSON line per mode.
Usage: python tools/score_bench.py [--device cuda] [--batch 65536]
"""
import argparse
import json
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import numpy as np
import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--batch", type=int, default=65536)
    ap.add_argument("--rows", type=int, default=1 << 20)
    ap.add_argument("--n-cat", type=int, default=26)
    ap.add_argument("--n-dense", type=int, default=200)
    ap.add_argument("--vocab", type=int, default=100000)
    ap.add_argument("--embed-dim", type=int, default=64)
    args = ap.parse_args()

    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.train.export import export_model
    from shifu_amd.serve import ShifuScorer

    torch.manual_seed(11)
    model = WideDeep(args.n_dense, [args.vocab] * args.n_cat, args.embed_dim,
                     [1024, 512, 256], ["relu"] * 3, seed=11)
    with tempfile.TemporaryDirectory() as td:
        export_model(model, td)
        sc = ShifuScorer()
        sc.init(os.path.join(td, "GenericModelConfig.json"),
                device=args.device)

        rng = np.random.default_rng(3)
        dense = rng.standard_normal((args.rows, args.n_dense), dtype=np.float32)
        cats = rng.integers(0, args.vocab, (args.rows, args.n_cat),
                            dtype=np.int64)

        # batched scoring
        n = 0
        # warmup
        sc.compute_batch(dense[:args.batch], cats[:args.batch])
        if args.device == "cuda":
            torch.cuda.synchronize()
        t0 = time.time()
        for s in range(0, args.rows, args.batch):
            e = min(s + args.batch, args.rows)
            p = sc.compute_batch(dense[s:e], cats[s:e])
            n += len(p)
        if args.device == "cuda":
            torch.cuda.synchronize()
        dt = time.time() - t0
        print(json.dumps({
            "metric": "scoring_rows_per_sec", "value": n / dt, "unit": "rows/s",
            "mode": "compute_batch", "device": args.device,
            "batch": args.batch, "rows": n, "data": "synthetic",
            "config": {"model": f"wide_deep[{args.n_cat}x{args.vocab}vocab"
                                f"*{args.embed_dim}d+{args.n_dense}dense,"
                                f"tower[1024, 512, 256]]"}}), flush=True)

        # per-row latency (Java Computable call pattern)
        k = 2000
        rows1 = [np.concatenate([dense[i], cats[i].astype(np.float64)])
                 for i in range(k)]
        sc.compute(rows1[0])
        t0 = time.time()
        for i in range(k):
            sc.compute(rows1[i])
        dt = time.time() - t0
        print(json.dumps({
            "metric": "scoring_row_latency_us", "value": dt / k * 1e6,
            "unit": "us/row", "mode": "compute", "device": args.device,
            "rows": k, "higher_is_better": False}), flush=True)


if __name__ == "__main__":
    main()
