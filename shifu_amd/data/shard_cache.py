"""Parsed-shard cache: restart-from-checkpoint without re-parsing the CSVs.

The launcher recovers from a dead rank by restarting the whole job from the
latest checkpoint (parallel/launcher.py — the single-node successor of the
reference's hot-spare promotion, TensorflowApplicationMaster.java:390-426).
At the 100M-row config the dominant restart cost is not the checkpoint load
but re-parsing the gzip CSVs (~minutes); this cache makes a restart's data
path a single mmap-fast tensor load.

Keying: file paths + (size, mtime_ns) of every input + the full column
selection + split parameters + (rank, world).  Any change invalidates the
entry; writes are atomic (tmp + rename), so a killed writer never leaves a
corrupt cache.
"""
from __future__ import annotations

import hashlib
import json
import os
from typing import Callable, Optional, Sequence, Tuple

import numpy as np
import torch

from shifu_amd.data.csv_loader import TabularDataset


def _key(files: Sequence[str], spec: dict) -> str:
    meta = []
    for f in sorted(files):
        st = os.stat(f)
        meta.append((os.path.abspath(f), st.st_size, st.st_mtime_ns))
    blob = json.dumps([meta, spec], sort_keys=True).encode()
    return hashlib.sha256(blob).hexdigest()[:20]


def _pack(ds: TabularDataset) -> dict:
    return {"dense": ds.dense, "cats": ds.cats,
            "target": ds.target, "weight": ds.weight}


def _unpack(d: dict) -> TabularDataset:
    return TabularDataset(d["dense"], d["cats"], d["target"], d["weight"])


def load_split_cached(cache_dir: Optional[str], files: Sequence[str],
                      spec: dict, rank: int, world: int,
                      build: Callable[[], Tuple[TabularDataset, TabularDataset]]
                      ) -> Tuple[TabularDataset, TabularDataset, bool]:
    """(train, valid, from_cache).  `build()` parses when there is no valid
    cache entry; cache_dir=None disables caching entirely."""
    if not cache_dir:
        t, v = build()
        return t, v, False
    try:
        key = _key(files, dict(spec, rank=rank, world=world))
    except OSError:
        t, v = build()
        return t, v, False
    path = os.path.join(cache_dir, f"shard-{rank}of{world}-{key}.pt")
    if os.path.exists(path):
        try:
            blob = torch.load(path, map_location="cpu", weights_only=False)
            return _unpack(blob["train"]), _unpack(blob["valid"]), True
        except Exception:
            pass   # unreadable entry: fall through to a fresh parse
    train, valid = build()
    try:
        os.makedirs(cache_dir, exist_ok=True)
        # prune this shard's entries under older keys (changed data/spec):
        # they can never be read again and a 100M-row entry is tens of GB
        prefix = f"shard-{rank}of{world}-"
        for name in os.listdir(cache_dir):
            if name.startswith(prefix) and name != os.path.basename(path):
                try:
                    os.remove(os.path.join(cache_dir, name))
                except OSError:
                    pass
        tmp = path + f".tmp{os.getpid()}"
        torch.save({"train": _pack(train), "valid": _pack(valid)}, tmp)
        os.replace(tmp, path)
    except OSError:
        pass       # cache is best-effort
    return train, valid, False
