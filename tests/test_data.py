"""CSV ingest + synthetic generator + sharding tests (reference semantics:
ssgd_monitor.py:348-454 load_data; TrainingDataSet.java:55-89 splitter)."""
import gzip
import os

import numpy as np
import pytest

from shifu_amd.data.csv_loader import (TabularDataset, count_total_rows,
                                       list_training_files, load_csv_files)
from shifu_amd.data.sharding import shard_files, shard_rows
from shifu_amd.data.synthetic import generate_synthetic_csv, synthetic_arrays


def test_load_gzip_csv(tmp_path):
    paths = generate_synthetic_csv(str(tmp_path), n_rows=200, n_dense=5,
                                   vocab_sizes=[10], n_files=2, seed=7)
    assert all(p.endswith(".gz") for p in paths)
    ds = load_csv_files(paths, selected_numeric=[2, 3, 4, 5, 6],
                        selected_categorical=[7], target_column=0, weight_column=1)
    assert len(ds) == 200
    assert ds.dense.shape == (200, 5)
    assert ds.cats.shape == (200, 1)
    assert ds.weight.min() >= 0
    assert set(np.unique(ds.target)) <= {0.0, 1.0}
    assert ds.pos_count + ds.neg_count == 200


def test_negative_weight_coerced(tmp_path):
    p = tmp_path / "x.csv"
    p.write_text("1|-2.5|0.1\n0|0.5|0.2\n")
    ds = load_csv_files([str(p)], selected_numeric=[2], target_column=0, weight_column=1)
    assert ds.weight[0] == 1.0  # negative coerced (ssgd_monitor.py:412-419)
    assert ds.weight[1] == 0.5


def test_malformed_rows_skipped(tmp_path):
    p = tmp_path / "x.csv"
    p.write_text("target|weight|a\n1|1.0|0.5\nnot|a|row\n0|1.0|0.25\n")
    ds = load_csv_files([str(p)], selected_numeric=[2], target_column=0, weight_column=1)
    assert len(ds) == 2


def test_no_weight_column(tmp_path):
    p = tmp_path / "x.csv"
    p.write_text("1|0.1\n0|0.2\n")
    ds = load_csv_files([str(p)], selected_numeric=[1], target_column=0, weight_column=-1)
    assert np.all(ds.weight == 1.0)


def test_split_deterministic():
    dense, cats, target, weight = synthetic_arrays(1000, 4, seed=3)
    ds = TabularDataset(dense, cats, target, weight)
    t1, v1 = ds.split(0.25, seed=42)
    t2, v2 = ds.split(0.25, seed=42)
    assert len(t1) == len(t2) and len(v1) == len(v2)
    assert np.allclose(t1.dense, t2.dense)
    assert abs(len(v1) / len(ds) - 0.25) < 0.05


def test_count_total_rows(tmp_path):
    paths = generate_synthetic_csv(str(tmp_path), n_rows=123, n_dense=2, n_files=3, seed=1)
    assert count_total_rows(paths) == 123


def test_list_training_files_skips_hidden(tmp_path):
    (tmp_path / "part-0.csv").write_text("x")
    (tmp_path / ".hidden").write_text("x")
    (tmp_path / "_SUCCESS").write_text("x")
    files = list_training_files(str(tmp_path))
    assert [os.path.basename(f) for f in files] == ["part-0.csv"]


def test_shard_files_round_robin():
    paths = [f"f{i}" for i in range(10)]
    shards = [shard_files(paths, r, 4) for r in range(4)]
    assert sorted(sum(shards, [])) == sorted(paths)
    assert shards[0] == ["f0", "f4", "f8"]


def test_shard_files_strict_raises():
    with pytest.raises(ValueError):
        shard_files(["a"], 0, 2, strict=True)  # TrainingDataSet.java:84-86 behavior


def test_shard_rows_cover():
    spans = [shard_rows(103, r, 8) for r in range(8)]
    assert spans[0][0] == 0 and spans[-1][1] == 103
    covered = sum(e - s for s, e in spans)
    assert covered == 103
    sizes = [e - s for s, e in spans]
    assert max(sizes) - min(sizes) <= 1


def test_shard_cache_roundtrip_and_invalidation(tmp_path):
    """data/shard_cache.py: second load comes from cache with identical
    arrays; touching an input invalidates the entry."""
    import os
    import time
    import numpy as np
    from shifu_amd.data.synthetic import generate_synthetic_csv
    from shifu_amd.data.csv_loader import load_csv_files, list_training_files
    from shifu_amd.data.shard_cache import load_split_cached

    data_dir = tmp_path / "data"
    generate_synthetic_csv(str(data_dir), 300, 5, (11,), seed=3, n_files=1)
    files = list_training_files([str(data_dir)])
    spec = {"num": [2, 3, 4, 5, 6], "cat": [7], "target": 0, "weight": 1,
            "valid": 0.2, "seed": 1}

    calls = {"n": 0}

    def build():
        calls["n"] += 1
        full = load_csv_files(files, [2, 3, 4, 5, 6], [7], 0, 1, "|")
        return full.split(0.2, seed=1)

    cache = str(tmp_path / "cache")
    t1, v1, c1 = load_split_cached(cache, files, spec, 0, 1, build)
    t2, v2, c2 = load_split_cached(cache, files, spec, 0, 1, build)
    assert not c1 and c2 and calls["n"] == 1
    assert np.array_equal(t1.dense, t2.dense)
    assert np.array_equal(t1.target, t2.target)
    assert np.array_equal(v1.dense, v2.dense)

    # different rank -> different entry
    _, _, c3 = load_split_cached(cache, files, spec, 1, 2, build)
    assert not c3 and calls["n"] == 2

    # touching the input invalidates
    time.sleep(0.01)
    os.utime(files[0])
    _, _, c4 = load_split_cached(cache, files, spec, 0, 1, build)
    assert not c4 and calls["n"] == 3

    # cache disabled
    _, _, c5 = load_split_cached(None, files, spec, 0, 1, build)
    assert not c5 and calls["n"] == 4


def test_shard_cache_prunes_stale_keys(tmp_path):
    """Writing a new entry for a (rank, world) removes that shard's
    older-key entries (they are unreadable forever and can be tens of GB)."""
    import os
    import time
    from shifu_amd.data.synthetic import generate_synthetic_csv
    from shifu_amd.data.csv_loader import load_csv_files, list_training_files
    from shifu_amd.data.shard_cache import load_split_cached

    data_dir = tmp_path / "data"
    generate_synthetic_csv(str(data_dir), 200, 4, (9,), seed=4, n_files=1)
    files = list_training_files([str(data_dir)])
    spec = {"num": [2, 3, 4, 5], "cat": [6], "target": 0, "weight": 1,
            "valid": 0.2, "seed": 1}

    def build():
        full = load_csv_files(files, [2, 3, 4, 5], [6], 0, 1, "|")
        return full.split(0.2, seed=1)

    cache = str(tmp_path / "cache")
    load_split_cached(cache, files, spec, 0, 1, build)
    assert len(os.listdir(cache)) == 1
    time.sleep(0.01)
    os.utime(files[0])                       # new key for the same shard
    load_split_cached(cache, files, spec, 0, 1, build)
    entries = [n for n in os.listdir(cache) if n.startswith("shard-0of1-")]
    assert len(entries) == 1, f"stale entry not pruned: {entries}"
