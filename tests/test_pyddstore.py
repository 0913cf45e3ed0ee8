"""Reference-API compatibility: the ``pyddstore.PyDDStore`` surface
(reference src/pyddstore.pyx:58-131) on NumPy arrays."""
import numpy as np
import pytest

import pyddstore


@pytest.fixture
def store():
    s = pyddstore.PyDDStore(device="cpu")
    yield s
    s.free()


def test_signature_surface(store):
    for m in ("add", "get", "init", "update", "epoch_begin", "epoch_end", "free"):
        assert hasattr(store, m)


def test_add_get_numpy(store):
    arr = np.random.rand(64, 8)
    store.add("t", arr)
    out = np.zeros((5, 8))
    store.epoch_begin()
    store.get("t", out, start=10)
    store.epoch_end()
    assert np.array_equal(out, arr[10:15])


def test_init_update_get(store):
    store.init("u", 16, 4, itemsize=8)
    store.update("u", np.full((4, 4), 3.0), offset=2)
    out = np.zeros((16, 4))
    store.get("u", out, 0)
    assert (out[2:6] == 3).all() and (out[:2] == 0).all() and (out[6:] == 0).all()


@pytest.mark.parametrize(
    "dtype", [np.int32, np.int64, np.uint8, np.float32, np.float64, np.bool_]
)
def test_six_reference_dtypes(store, dtype):
    if dtype == np.bool_:
        arr = (np.arange(20) % 2 == 0).reshape(10, 2)
    else:
        arr = np.arange(20, dtype=dtype).reshape(10, 2)
    store.add(f"d_{np.dtype(dtype).name}", arr)
    out = np.zeros((10, 2), dtype=dtype)
    store.get(f"d_{np.dtype(dtype).name}", out, 0)
    assert np.array_equal(out, arr)


def test_noncontiguous_rejected(store):
    arr = np.random.rand(8, 8)[:, ::2]
    with pytest.raises(AssertionError, match="contiguous"):
        store.add("nc", arr)


def test_query_extension(store):
    store.add("q", np.zeros((4, 2), dtype=np.float32))
    q = store.query("q")
    assert q["nrows_total"] == 4 and q["disp"] == 2
