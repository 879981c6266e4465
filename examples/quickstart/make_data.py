#!/usr/bin/env python3
"""Generate the quickstart's synthetic normalized dataset (gzip '|'-CSV,
Shifu layout: target col 0, weight col 1, 6 dense, 2 categorical)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))
from shifu_amd.data.synthetic import generate_synthetic_csv

out = sys.argv[1] if len(sys.argv) > 1 else "./data"
generate_synthetic_csv(out, n_rows=20000, n_dense=6, vocab_sizes=[50, 80],
                       n_files=4, seed=7)
print(f"wrote 4 csv.gz shards (20k rows) under {out}")
