"""Deterministic synthetic tabular data.

Two forms:
* generate_synthetic_csv — writes Shifu-style '|'-delimited csv(.gz) files so
  the full ingest path (csv_loader) is exercised end-to-end in tests;
* synthetic_tensors — generates the same distribution directly as tensors for
  benchmarking (bench.py: "data": "synthetic"), skipping file IO.

The label is a noisy linear-logit function of the features so a trained model
shows a real AUC > 0.5 (AUC-parity checks in BASELINE.md).
"""
from __future__ import annotations

import gzip
import os
from typing import List, Optional, Sequence, Tuple

import numpy as np


def _logits(rng: np.random.Generator, dense: np.ndarray, cats: np.ndarray,
            vocab_sizes: Sequence[int]) -> np.ndarray:
    n, f_num = dense.shape
    w = rng.standard_normal(f_num).astype(np.float32) / max(np.sqrt(f_num), 1.0)
    logit = dense @ w
    for j, v in enumerate(vocab_sizes):
        # per-category contribution: a cheap deterministic hash -> [-0.5, 0.5]
        logit += ((cats[:, j] * 2654435761 % 1000) / 1000.0 - 0.5).astype(np.float32)
    logit += 0.5 * rng.standard_normal(n).astype(np.float32)
    return logit


def synthetic_arrays(
    n_rows: int,
    n_dense: int,
    vocab_sizes: Sequence[int] = (),
    seed: int = 1234,
    weighted: bool = True,
) -> Tuple[np.ndarray, np.ndarray, np.ndarray, np.ndarray]:
    """Returns (dense [N,Fn] f32, cats [N,Fc] i64, target [N] f32, weight [N] f32)."""
    rng = np.random.default_rng(seed)
    dense = rng.standard_normal((n_rows, n_dense)).astype(np.float32)
    cats = np.zeros((n_rows, len(vocab_sizes)), dtype=np.int64)
    for j, v in enumerate(vocab_sizes):
        # zipf-ish skew: hot rows exist, like real categorical traffic
        u = rng.random(n_rows)
        cats[:, j] = np.minimum((u ** 2.0 * v).astype(np.int64), v - 1)
    logit = _logits(rng, dense, cats, vocab_sizes)
    prob = 1.0 / (1.0 + np.exp(-logit))
    target = (rng.random(n_rows) < prob).astype(np.float32)
    weight = (rng.random(n_rows) * 2.0).astype(np.float32) if weighted \
        else np.ones(n_rows, dtype=np.float32)
    return dense, cats, target, weight


def generate_synthetic_csv(
    out_dir: str,
    n_rows: int,
    n_dense: int,
    vocab_sizes: Sequence[int] = (),
    n_files: int = 1,
    seed: int = 1234,
    gz: bool = True,
    delimiter: str = "|",
    weighted: bool = True,
) -> List[str]:
    """Write synthetic rows as Shifu-normalized CSV.

    Column layout: [target, weight, dense..., cats...] — i.e. target_column=0,
    weight_column=1, numeric columns 2..2+n_dense-1, categorical after.
    """
    dense, cats, target, weight = synthetic_arrays(
        n_rows, n_dense, vocab_sizes, seed=seed, weighted=weighted)
    os.makedirs(out_dir, exist_ok=True)
    paths = []
    splits = np.array_split(np.arange(n_rows), n_files)
    for fi, idx in enumerate(splits):
        name = f"part-{fi:05d}.csv" + (".gz" if gz else "")
        path = os.path.join(out_dir, name)
        opener = gzip.open if gz else open
        with opener(path, "wt") as f:
            for i in idx:
                fields = [f"{target[i]:.0f}", f"{weight[i]:.6f}"]
                fields += [f"{x:.6f}" for x in dense[i]]
                fields += [str(int(x)) for x in cats[i]]
                f.write(delimiter.join(fields) + "\n")
        paths.append(path)
    return paths


def synthetic_tensors(
    n_rows: int,
    n_dense: int,
    vocab_sizes: Sequence[int] = (),
    seed: int = 1234,
    device: str = "cpu",
    dense_dtype=None,
):
    """Tensor form for bench.py: (dense, cats, target, weight) torch tensors."""
    import torch
    dense, cats, target, weight = synthetic_arrays(n_rows, n_dense, vocab_sizes, seed=seed)
    td = torch.from_numpy(dense)
    if dense_dtype is not None:
        td = td.to(dense_dtype)
    return (td.to(device),
            torch.from_numpy(cats).to(device),
            torch.from_numpy(target).to(device),
            torch.from_numpy(weight).to(device))
