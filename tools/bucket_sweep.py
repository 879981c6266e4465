#!/usr/bin/env python3
"""Multi-GPU knob sweep harness (round-2 tool; needs an N-GPU box).

Runs bench.py under torchrun for each (bucket_mb, emb_mode) combination and
prints one result line per run.  Usage on an 8-GPU node:

    python tools/bucket_sweep.py --gpus 8 [--buckets 32,64,128,256] \
        [--emb-modes ep,dp] [--steps 30] [--warmup 8]

The all-reduce bucket size trades hook-overlap granularity against per-link
xGMI message efficiency (7x ~153 GB/s point-to-point links); the shipped
default (128 MB) was chosen analytically — this measures it.
"""
import argparse
import json
import subprocess
import sys


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=8)
    ap.add_argument("--buckets", default="32,64,128,256")
    ap.add_argument("--emb-modes", default="ep")
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--port", type=int, default=29581)
    args = ap.parse_args()

    for mode in args.emb_modes.split(","):
        for mb in (int(x) for x in args.buckets.split(",")):
            cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
                   "--nproc-per-node", str(args.gpus),
                   "--master-addr", "127.0.0.1", "--master-port", str(args.port),
                   "bench.py", "--gpus", str(args.gpus),
                   "--steps", str(args.steps), "--warmup", str(args.warmup),
                   "--bucket-mb", str(mb), "--emb-mode", mode]
            r = subprocess.run(cmd, capture_output=True, text=True, timeout=1800)
            line = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
            if r.returncode != 0 or not line:
                print(f"bucket={mb} emb={mode}: FAILED\n{r.stderr[-800:]}",
                      flush=True)
                continue
            d = json.loads(line[-1])
            print(f"bucket={mb:>4} emb={mode}: {d['value']:,.0f} samples/s "
                  f"({d['ms_per_step']:.3f} ms/step)", flush=True)


if __name__ == "__main__":
    main()
