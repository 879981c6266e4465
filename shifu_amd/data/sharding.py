"""Data sharding across ranks.

Successor of the reference's file-level round-robin splitter
(reference: appmaster/TrainingDataSet.java:55-89): files are round-robined
across workers, '.'/'_' files skipped, and the reference throws when there are
fewer files than workers (TrainingDataSet.java:84-86).  We keep file-level
sharding for multi-file inputs but add row-range sharding so a single big
file (or in-memory synthetic data) still splits evenly across 8 ranks.
"""
from __future__ import annotations

from typing import List, Sequence, Tuple


def shard_files(paths: Sequence[str], rank: int, world_size: int,
                strict: bool = False) -> List[str]:
    """Round-robin file assignment: file i -> rank i % world_size."""
    if strict and len(paths) < world_size:
        # reference behavior: "Number of training files is less than workers"
        raise ValueError(
            f"number of training files ({len(paths)}) is less than workers ({world_size})")
    return [p for i, p in enumerate(paths) if i % world_size == rank]


def shard_rows(n_rows: int, rank: int, world_size: int) -> Tuple[int, int]:
    """Contiguous row-range [start, end) for this rank; remainder spread over
    the first ranks so sizes differ by at most 1."""
    base = n_rows // world_size
    rem = n_rows % world_size
    start = rank * base + min(rank, rem)
    end = start + base + (1 if rank < rem else 0)
    return start, end
