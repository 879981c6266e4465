#!/usr/bin/env python3
"""Sweep the all-reduce bucket size on the real bench at a given world size.

The 128 MB default was chosen from xGMI topology math (7 p2p links, ring
all-reduce per-link bound — ARCHITECTURE §2); this sweeps it empirically on
hardware.  Run on a multi-GPU node:

    python tools/bucket_sweep.py --gpus 8 [--buckets 32,64,128,256]
        [--steps 30 --warmup 8]

Each point launches bench.py via torch.distributed.run (one rank per GPU)
and reports the whole-job samples/s.  Also usable at --gpus 1 (bucketing is
then inert — a sanity floor) and on CPU/gloo for plumbing tests.
"""
import argparse
import json
import subprocess
import sys
import os

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_point(gpus: int, bucket_mb: int, steps: int, warmup: int,
              extra: list) -> float:
    cmd = [sys.executable]
    if gpus > 1:
        cmd += ["-m", "torch.distributed.run", "--nnodes=1",
                "--nproc-per-node", str(gpus),
                "--master-addr", "127.0.0.1", "--master-port", "29741"]
    cmd += [os.path.join(ROOT, "bench.py"), "--steps", str(steps),
            "--warmup", str(warmup), "--bucket-mb", str(bucket_mb)] + extra
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=1800)
    if out.returncode != 0:
        print(out.stderr[-2000:], file=sys.stderr)
        raise SystemExit(f"bench failed at bucket_mb={bucket_mb}")
    line = [l for l in out.stdout.strip().splitlines() if l.startswith("{")][-1]
    return float(json.loads(line)["value"])


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=8)
    ap.add_argument("--buckets", default="32,64,128,256")
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--extra", default="",
                    help="extra bench.py flags, e.g. '--batch 16384'")
    args = ap.parse_args()
    extra = args.extra.split() if args.extra else []
    results = {}
    for mb in [int(b) for b in args.buckets.split(",")]:
        v = run_point(args.gpus, mb, args.steps, args.warmup, extra)
        results[mb] = v
        print(f"bucket_mb={mb:<5d} {v/1e6:8.2f}M samples/s", flush=True)
    best = max(results, key=results.get)
    print(json.dumps({"metric": "bucket_sweep", "gpus": args.gpus,
                      "results": results, "best_bucket_mb": best}))


if __name__ == "__main__":
    main()
