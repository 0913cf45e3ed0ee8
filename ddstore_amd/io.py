"""Dataset ingest helpers: file -> sharded store.

The reference leaves loading to the caller (HydraGNN feeds preprocessed
pickles; the VAE example downloads MNIST). These helpers do the standard
thing for the rebuild: every rank memory-maps the SAME dataset file and
ingests only its contiguous slice -- no rank ever materializes the full
dataset, so stores larger than host RAM load fine.
"""
from __future__ import annotations

from typing import Optional, Sequence, Union

import numpy as np

from .distdataset import nsplit
from .store import DDStore


def _my_slice(total: int, rank: int, size: int):
    counts = nsplit(total, size)
    lo = sum(counts[:rank])
    return lo, lo + counts[rank]


def add_from_npy(store: DDStore, name: str, path: str) -> None:
    """Register variable ``name`` from a .npy file (memory-mapped; each rank
    reads only rows [lo, hi) of its own shard)."""
    arr = np.load(path, mmap_mode="r")
    lo, hi = _my_slice(arr.shape[0], store.rank, store.size)
    shard = np.ascontiguousarray(arr[lo:hi]).reshape(hi - lo, -1)
    store.add(name, shard)


def add_from_memmap(
    store: DDStore,
    name: str,
    path: str,
    dtype: Union[str, np.dtype],
    row_shape: Sequence[int],
    nrows: Optional[int] = None,
) -> None:
    """Register from a raw binary file of ``nrows`` rows of ``row_shape``
    elements of ``dtype`` (row-major). ``nrows=None`` infers from the file
    size."""
    dtype = np.dtype(dtype)
    row_elems = int(np.prod(row_shape)) if len(row_shape) else 1
    if nrows is None:
        import os

        sz = os.path.getsize(path)
        if sz % (row_elems * dtype.itemsize):
            raise ValueError("ddstore io: file size is not a whole number of rows")
        nrows = sz // (row_elems * dtype.itemsize)
    mm = np.memmap(path, dtype=dtype, mode="r", shape=(int(nrows), row_elems))
    lo, hi = _my_slice(int(nrows), store.rank, store.size)
    store.add(name, np.ascontiguousarray(mm[lo:hi]))


def add_csr_from_npy(store: DDStore, name: str, values_path: str,
                     lengths_path: str) -> None:
    """Register a CSR variable from two .npy files: per-sample ``lengths``
    and the concatenated ``values``. Each rank ingests the element range of
    its contiguous sample slice."""
    lengths = np.load(lengths_path)
    values = np.load(values_path, mmap_mode="r")
    lo, hi = _my_slice(lengths.shape[0], store.rank, store.size)
    off = np.zeros(lengths.shape[0] + 1, dtype=np.int64)
    np.cumsum(lengths, out=off[1:])
    vals = np.ascontiguousarray(values[off[lo] : off[hi]])
    store.add_csr(name, vals, lengths[lo:hi])
