"""Native C++ CSV reader: parity with the Python parser + throughput sanity."""
import time

import numpy as np
import pytest

from shifu_amd.data.csv_loader import load_csv_files
from shifu_amd.data.synthetic import generate_synthetic_csv
from shifu_amd.io import count_rows_native, load_csv_native, native_io

requires_native = pytest.mark.skipif(native_io() is None,
                                     reason="_shifu_io not built")


@requires_native
def test_native_matches_python(tmp_path):
    paths = generate_synthetic_csv(str(tmp_path), n_rows=500, n_dense=6,
                                   vocab_sizes=[40, 10], n_files=3, seed=3)
    kw = dict(selected_numeric=[2, 3, 4, 5, 6, 7], selected_categorical=[8, 9],
              target_column=0, weight_column=1)
    a = load_csv_files(paths, **kw)
    b = load_csv_native(paths, **kw)
    assert len(a) == len(b) == 500
    assert np.allclose(a.dense, b.dense, atol=1e-6)
    assert np.array_equal(a.cats, b.cats)
    assert np.array_equal(a.target, b.target)
    assert np.allclose(a.weight, b.weight, atol=1e-6)


@requires_native
def test_native_semantics_edge_cases(tmp_path):
    p = tmp_path / "x.csv"
    p.write_text("header|noise|x\n"        # malformed -> skipped
                 "1|-2.5|0.5\n"            # negative weight -> 1.0
                 "0|0.25|0.75\n"
                 "1|abc|0.1\n"             # unparseable weight -> 1.0
                 "0|1.0\n")                # missing selected col -> skipped
    ds = load_csv_native([str(p)], selected_numeric=[2], target_column=0,
                         weight_column=1)
    assert len(ds) == 3
    assert ds.weight.tolist() == [1.0, 0.25, 1.0]
    assert ds.dense.reshape(-1).tolist() == pytest.approx([0.5, 0.75, 0.1])


@requires_native
def test_native_count_rows(tmp_path):
    paths = generate_synthetic_csv(str(tmp_path), n_rows=321, n_dense=2,
                                   n_files=4, seed=5)
    assert count_rows_native(paths) == 321


@requires_native
def test_native_throughput(tmp_path):
    """Native reader must beat the Python parser by a wide margin (the whole
    point — config 5 ingest)."""
    paths = generate_synthetic_csv(str(tmp_path), n_rows=60000, n_dense=20,
                                   n_files=4, seed=6)
    kw = dict(selected_numeric=list(range(2, 22)), target_column=0, weight_column=1)
    t0 = time.time()
    a = load_csv_native(paths, **kw)
    t_native = time.time() - t0
    t0 = time.time()
    b = load_csv_files(paths, **kw)
    t_python = time.time() - t0
    assert len(a) == len(b) == 60000
    assert t_native < t_python / 3, f"native {t_native:.2f}s vs python {t_python:.2f}s"


def test_python_fallback_when_ext_missing(tmp_path, monkeypatch):
    """With the extension forced absent, load_csv_native must transparently
    fall back to the pure-Python parser with identical results."""
    import shifu_amd.io as io_mod
    from shifu_amd.data.synthetic import generate_synthetic_csv
    paths = generate_synthetic_csv(str(tmp_path), n_rows=200, n_dense=3,
                                   vocab_sizes=[9], n_files=2, seed=4)
    cols = dict(selected_numeric=[2, 3, 4], selected_categorical=[5],
                target_column=0, weight_column=1)
    native = load_csv_native(paths, **cols)
    monkeypatch.setattr(io_mod, "_EXT", None)
    monkeypatch.setattr(io_mod, "_TRIED", True)
    fallback = io_mod.load_csv_native(paths, **cols)
    assert io_mod.native_io() is None
    assert len(fallback) == len(native)
    import numpy as np
    np.testing.assert_allclose(fallback.dense, native.dense, atol=1e-6)
    np.testing.assert_array_equal(fallback.cats, native.cats)
    np.testing.assert_allclose(fallback.target, native.target, atol=1e-6)
    np.testing.assert_allclose(fallback.weight, native.weight, atol=1e-6)


def test_no_target_layout(tmp_path):
    """target_column=-1 (scoring-only datasets): rows parse with zero
    targets, column 0 stays a plain feature — native and python loaders
    agree (the round-1 score.py raw layout reused column 0 as target)."""
    import numpy as np
    from shifu_amd.data.csv_loader import load_csv_files
    from shifu_amd.io import load_csv_native
    p = tmp_path / "raw.csv"
    p.write_text("0.5|1.5|7\n-0.25|2.5|3\n")
    for loader in (load_csv_files, load_csv_native):
        ds = loader([str(p)], [0, 1], [2], -1, -1, "|")
        assert len(ds) == 2
        assert np.allclose(ds.dense[:, 0], [0.5, -0.25])
        assert np.all(ds.target == 0.0)
        assert np.all(ds.weight == 1.0)
