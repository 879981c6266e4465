"""Gzip'd '|'-delimited normalized-CSV ingest with Shifu semantics.

Reproduces the behavioral contract of the reference's load_data
(reference: shifu-tensorflow-on-yarn/src/main/resources/ssgd_monitor.py:348-454):

* input: one or more csv / csv.gz files of '|'-delimited floats (Shifu
  "normalized" data), no header;
* a target column (binary 0/1), an optional per-row sample-weight column
  (negative or unparseable weights coerced to 1.0, ssgd_monitor.py:412-419);
* selected feature columns by index;
* random train/valid split by validSetRate with a seeded RNG;
* pos/neg counts reported.

Unlike the reference (python lists, one float at a time), rows are parsed
into numpy arrays — the 100M-row config cannot materialize python lists
(SURVEY.md §7 item 6).  Categorical columns carry integer ids (embedding
indices); numeric columns carry normalized floats.
"""
from __future__ import annotations

import gzip
import io
import os
from dataclasses import dataclass
from typing import List, Optional, Sequence, Tuple

import numpy as np


@dataclass
class TabularDataset:
    """In-memory columnar dataset: dense floats + categorical ids + target + weight."""
    dense: np.ndarray          # [N, F_num] float32
    cats: np.ndarray           # [N, F_cat] int64 (empty -> shape [N, 0])
    target: np.ndarray         # [N] float32 (0/1)
    weight: np.ndarray         # [N] float32 (>0)

    def __len__(self) -> int:
        return self.target.shape[0]

    @property
    def num_dense(self) -> int:
        return self.dense.shape[1]

    @property
    def num_cat(self) -> int:
        return self.cats.shape[1]

    @property
    def pos_count(self) -> int:
        return int((self.target >= 0.5).sum())

    @property
    def neg_count(self) -> int:
        return len(self) - self.pos_count

    def subset(self, idx: np.ndarray) -> "TabularDataset":
        return TabularDataset(self.dense[idx], self.cats[idx],
                              self.target[idx], self.weight[idx])

    def split(self, valid_rate: float, seed: int) -> Tuple["TabularDataset", "TabularDataset"]:
        """Random train/valid split (ssgd_monitor.py: random() < validSetRate per row)."""
        n = len(self)
        rng = np.random.default_rng(seed)
        is_valid = rng.random(n) < valid_rate
        return self.subset(~is_valid), self.subset(is_valid)


def _open_maybe_gzip(path: str) -> io.TextIOBase:
    if path.endswith(".gz"):
        return io.TextIOWrapper(gzip.open(path, "rb"))
    return open(path, "r")


def load_csv_files(
    paths: Sequence[str],
    selected_numeric: Sequence[int],
    selected_categorical: Sequence[int] = (),
    target_column: int = 0,
    weight_column: int = -1,
    delimiter: str = "|",
) -> TabularDataset:
    """Parse csv(.gz) files into a TabularDataset.

    Column semantics follow the env contract: indices are
    0-based positions in the delimited row; weight_column==-1 means all-ones.
    Rows that fail to parse are skipped (the reference tolerates bad rows).
    """
    num_cols = list(selected_numeric)
    cat_cols = list(selected_categorical)
    dense_rows: List[np.ndarray] = []
    cat_rows: List[np.ndarray] = []
    targets: List[float] = []
    weights: List[float] = []

    for path in paths:
        with _open_maybe_gzip(path) as f:
            for line in f:
                line = line.rstrip("\n\r")
                if not line:
                    continue
                parts = line.split(delimiter)
                try:
                    # target_column < 0: no-target (scoring-only) layout
                    t = float(parts[target_column]) if target_column >= 0 else 0.0
                except (ValueError, IndexError):
                    continue  # skip header-ish / malformed rows
                if weight_column >= 0:
                    try:
                        w = float(parts[weight_column])
                    except (ValueError, IndexError):
                        w = 1.0
                    if w < 0.0:
                        w = 1.0  # negative weights coerced (ssgd_monitor.py:412-419)
                else:
                    w = 1.0
                try:
                    drow = np.array([float(parts[i]) for i in num_cols], dtype=np.float32)
                    crow = np.array([int(float(parts[i])) for i in cat_cols], dtype=np.int64)
                except (ValueError, IndexError):
                    continue
                dense_rows.append(drow)
                cat_rows.append(crow)
                targets.append(t)
                weights.append(w)

    n = len(targets)
    dense = (np.stack(dense_rows) if n else np.zeros((0, len(num_cols)), np.float32))
    cats = (np.stack(cat_rows) if n else np.zeros((0, len(cat_cols)), np.int64))
    if cats.ndim == 1:
        cats = cats.reshape(n, -1)
    return TabularDataset(
        dense=dense.astype(np.float32),
        cats=cats.astype(np.int64),
        target=np.asarray(targets, dtype=np.float32),
        weight=np.asarray(weights, dtype=np.float32),
    )


def count_total_rows(paths: Sequence[str]) -> int:
    """Total row count across files (successor of HdfsUtils.getFileLineCount,
    reference: util/HdfsUtils.java:143-175, which feeds TOTAL_TRAINING_DATA_NUMBER)."""
    total = 0
    for path in paths:
        with _open_maybe_gzip(path) as f:
            for line in f:
                if line.strip():
                    total += 1
    return total


def list_training_files(root_or_files) -> List[str]:
    """Expand a directory or list of paths into training files, skipping
    '.'/'_'-prefixed entries exactly like the reference's splitter
    (reference: appmaster/TrainingDataSet.java:69-71)."""
    if isinstance(root_or_files, str):
        root_or_files = [root_or_files]
    out: List[str] = []
    for p in root_or_files:
        if os.path.isdir(p):
            for name in sorted(os.listdir(p)):
                if name.startswith(".") or name.startswith("_"):
                    continue
                out.append(os.path.join(p, name))
        else:
            out.append(p)
    return out
