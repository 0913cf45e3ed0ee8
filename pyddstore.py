"""pyddstore -- drop-in compatible Python API for the MI355X-native store.

Same module name, class name, and method signatures as the reference binding
(reference: src/pyddstore.pyx:58-131): ``PyDDStore(comm, method=0)`` with
``add/get/init/update/epoch_begin/epoch_end/free`` operating on NumPy arrays,
including the same six dtypes (int32, int64, uint8, float32, float64, bool --
pyddstore.pyx:69-82; note the reference uses the removed ``np.bool`` alias,
pyddstore.pyx:79, fixed here). ``ddstore_width`` is additionally accepted in
the constructor, as the reference README documents but its binding omits
(README.md:71-77 vs pyddstore.pyx:61).

On a GPU node the shards live in HBM3E and NumPy I/O is staged through the
device; set ``DDSTORE_DEVICE=cpu`` (or pass ``device="cpu"``) for the pure
host path. ``comm`` may be an mpi4py communicator, a torch.distributed
group, or None (torch.distributed WORLD / single rank).
"""
from __future__ import annotations

from typing import Optional

import numpy as np

from ddstore_amd import DDStore

__all__ = ["PyDDStore"]


class PyDDStore:
    def __init__(self, comm=None, method: int = 0, ddstore_width: Optional[int] = None,
                 device=None):
        self._store = DDStore(comm, method=method, device=device,
                              ddstore_width=ddstore_width)

    # reference-parity surface ------------------------------------------------
    def add(self, name: str, arr: np.ndarray) -> None:
        assert arr.flags["C_CONTIGUOUS"], "array must be C-contiguous"
        self._store.add(name, arr)

    def get(self, name: str, arr: np.ndarray, start: int = 0) -> None:
        assert arr.flags["C_CONTIGUOUS"], "array must be C-contiguous"
        self._store.get(name, arr, start=start)

    def init(self, name: str, nrows: int, disp: int, itemsize: int = 1) -> None:
        self._store.init(name, nrows, disp, itemsize=itemsize)

    def update(self, name: str, arr: np.ndarray, offset: int = 0) -> None:
        assert arr.flags["C_CONTIGUOUS"], "array must be C-contiguous"
        self._store.update(name, arr, offset)

    def epoch_begin(self) -> None:
        self._store.epoch_begin()

    def epoch_end(self) -> None:
        self._store.epoch_end()

    def free(self) -> None:
        self._store.free()

    # conveniences beyond the reference --------------------------------------
    def query(self, name: str) -> dict:
        return self._store.query(name)

    @property
    def store(self) -> DDStore:
        return self._store
