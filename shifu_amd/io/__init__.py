"""Native IO: multithreaded gzip-CSV ingest (C++/zlib) with a pure-Python
fallback (shifu_amd/data/csv_loader.py) when the extension isn't built."""
from __future__ import annotations

from typing import Sequence

_EXT = None
_TRIED = False


def native_io():
    global _EXT, _TRIED
    if not _TRIED:
        _TRIED = True
        try:
            import torch  # noqa: F401  (loads libc10 the extension links against)
            from shifu_amd.io import _shifu_io  # type: ignore
            _EXT = _shifu_io
        except Exception:
            _EXT = None
    return _EXT


def load_csv_native(paths: Sequence[str], selected_numeric: Sequence[int],
                    selected_categorical: Sequence[int] = (),
                    target_column: int = 0, weight_column: int = -1,
                    delimiter: str = "|", nthreads: int = 16):
    """TabularDataset via the native reader; falls back to the Python parser."""
    from shifu_amd.data.csv_loader import TabularDataset, load_csv_files
    ext = native_io()
    if ext is None:
        return load_csv_files(paths, selected_numeric, selected_categorical,
                              target_column, weight_column, delimiter)
    dense, cats, target, weight = ext.load_csv(
        list(paths), list(selected_numeric), list(selected_categorical),
        int(target_column), int(weight_column), delimiter, int(nthreads))
    return TabularDataset(dense.numpy(), cats.numpy(), target.numpy(), weight.numpy())


def count_rows_native(paths: Sequence[str], nthreads: int = 16) -> int:
    ext = native_io()
    if ext is None:
        from shifu_amd.data.csv_loader import count_total_rows
        return count_total_rows(paths)
    return int(ext.count_rows(list(paths), int(nthreads)))
