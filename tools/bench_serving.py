#!/usr/bin/env python3
"""Serving-path benchmark: single-row latency and batch scoring throughput
over an exported Wide&Deep bundle (the shifu-tensorflow-eval successor).

Usage: python tools/bench_serving.py [--device cuda]
Prints one JSON line.
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
    ap.add_argument("--device", default="cuda" if torch.cuda.is_available() else "cpu")
    ap.add_argument("--rows", type=int, default=200_000)
    args = ap.parse_args()

    from shifu_amd.models.wide_deep import WideDeep
    from shifu_amd.train.export import export_model
    from shifu_amd.serve import ShifuScorer

    model = WideDeep(200, [100_000] * 26, 32, [512, 256], ["relu", "relu"], seed=1)
    with tempfile.TemporaryDirectory() as td:
        export_model(model, td)
        sc = ShifuScorer()
        sc.init(os.path.join(td, "GenericModelConfig.json"), device=args.device)

        rng = np.random.default_rng(0)
        row = list(rng.standard_normal(200)) + [int(x) for x in
                                                rng.integers(0, 100_000, 26)]
        # single-row latency
        for _ in range(10):
            sc.compute(row)
        t0 = time.time()
        n_lat = 200
        for _ in range(n_lat):
            sc.compute(row)
        lat_ms = (time.time() - t0) / n_lat * 1000.0

        # batch throughput
        dense = rng.standard_normal((args.rows, 200)).astype(np.float32)
        cats = rng.integers(0, 100_000, (args.rows, 26))
        sc.compute_batch(dense[:1000], cats[:1000])  # warmup
        if args.device == "cuda":
            torch.cuda.synchronize()
        t0 = time.time()
        scores = sc.compute_batch(dense, cats)
        el = time.time() - t0

    print(json.dumps({
        "metric": "serving", "device": args.device,
        "single_row_latency_ms": lat_ms,
        "batch_rows_per_sec": args.rows / el,
        "batch_rows": args.rows,
        "score_range_ok": bool((scores >= 0).all() and (scores <= 1).all()),
    }), flush=True)


if __name__ == "__main__":
    main()
