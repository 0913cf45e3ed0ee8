"""Single-rank CPU store tests: API surface, dtype dispatch, error semantics.

Mirrors the reference's self-verifying synthetic-data philosophy
plus the error paths the reference defines but never tests (ddstore.hpp:81-82,
:202-203, :210-214; ddstore.cxx:57, :71).
"""
import numpy as np
import pytest
import torch

from ddstore_amd import DDStore

# the six reference dtypes (pyddstore.pyx:69-82) plus the new half types
REF_DTYPES = [np.int32, np.int64, np.uint8, np.float32, np.float64, np.bool_]


@pytest.fixture
def store():
    s = DDStore(device="cpu")
    yield s
    s.free()


@pytest.mark.parametrize("dtype", REF_DTYPES)
def test_roundtrip_dtypes(store, dtype):
    if dtype == np.bool_:
        arr = (np.arange(40) % 3 == 0).reshape(10, 4)
    else:
        arr = np.arange(40, dtype=dtype).reshape(10, 4)
    store.add(f"v{np.dtype(dtype).name}", arr)
    out = np.zeros((4, 4), dtype=dtype)
    store.get(f"v{np.dtype(dtype).name}", out, start=3)
    assert np.array_equal(out, arr[3:7])


@pytest.mark.parametrize("tdtype", [torch.float16, torch.bfloat16])
def test_half_dtypes(store, tdtype):
    arr = torch.arange(24, dtype=tdtype).reshape(6, 4)
    store.add("h", arr)
    out = store.get_batch("h", [5, 0, 2])
    assert out.dtype == tdtype
    assert torch.equal(out, arr[[5, 0, 2]])


def test_get_batch_and_cast(store):
    arr = np.random.rand(32, 8).astype(np.float32)
    store.add("x", arr)
    idx = [31, 0, 7, 7, 16]
    out = store.get_batch("x", idx)
    assert np.array_equal(out.numpy(), arr[idx])
    out64 = store.get_batch("x", idx, dtype=torch.float64)
    assert out64.dtype == torch.float64
    assert np.allclose(out64.numpy(), arr[idx].astype(np.float64))


def test_1d_array_disp1(store):
    # reference: 1-D arrays get disp=1, element-addressed (pyddstore.pyx:67-68)
    arr = np.arange(100, dtype=np.float64)
    store.add("flat", arr)
    q = store.query("flat")
    assert q["disp"] == 1 and q["nrows_total"] == 100
    out = np.zeros(7, dtype=np.float64)
    store.get("flat", out, start=42)
    assert np.array_equal(out, arr[42:49])


def test_init_update(store):
    store.init("y", 10, 4, itemsize=8)
    out = np.zeros((10, 4), dtype=np.float64)
    store.get("y", out, 0)
    assert (out == 0).all()
    store.update("y", np.full((3, 4), 5.0), offset=7)
    store.get("y", out, 0)
    assert (out[7:] == 5).all() and (out[:7] == 0).all()
    # itemsize (not dtype) is what update checks -- reference ddstore.hpp:189-190
    store.update("y", np.arange(4, dtype=np.int64).reshape(1, 4), offset=0)
    with pytest.raises(RuntimeError, match="itemsize"):
        store.update("y", np.zeros((1, 4), dtype=np.float32))


def test_update_out_of_range(store):
    store.init("z", 4, 2, itemsize=4)
    with pytest.raises(RuntimeError, match="out of range"):
        store.update("z", np.zeros((2, 2), dtype=np.float32), offset=3)


def test_invalid_start(store):
    store.add("x", np.zeros((5, 2), dtype=np.float32))
    out = np.zeros((1, 2), dtype=np.float32)
    with pytest.raises(RuntimeError, match="Invalid start on target"):
        store.get("x", out, start=5)
    with pytest.raises(RuntimeError, match="Invalid start on target"):
        store.get("x", out, start=-1)
    out5 = np.zeros((3, 2), dtype=np.float32)
    with pytest.raises(RuntimeError, match="Invalid count on target"):
        store.get("x", out5, start=3)


def test_epoch_fsm(store):
    # reference throws on double-begin / end-without-begin (ddstore.cxx:57,:71)
    store.epoch_begin()
    with pytest.raises(RuntimeError, match="epoch already began"):
        store.epoch_begin()
    store.epoch_end()
    with pytest.raises(RuntimeError, match="epoch has not begun"):
        store.epoch_end()


def test_unknown_and_duplicate(store):
    with pytest.raises((RuntimeError, KeyError)):
        store.query("nope")
    store.add("dup", np.zeros((2, 2), dtype=np.float32))
    with pytest.raises(RuntimeError, match="already exists"):
        store.add("dup", np.zeros((2, 2), dtype=np.float32))


def test_get_dtype_itemsize_check(store):
    store.add("x", np.zeros((5, 2), dtype=np.float32))
    bad = np.zeros((2, 2), dtype=np.float64)
    with pytest.raises(RuntimeError, match="itemsize"):
        store.get("x", bad, 0)


def test_csr_roundtrip(store):
    lengths = [3, 0, 5, 2]
    vals = np.arange(10 * 2, dtype=np.float32).reshape(10, 2)
    store.add_csr("c", vals, lengths)
    v, off = store.get_csr("c", [2, 0, 1, 3])
    assert off.tolist() == [0, 5, 8, 8, 10]
    assert np.array_equal(v[0:5].numpy(), vals[3:8])
    assert np.array_equal(v[5:8].numpy(), vals[0:3])
    assert np.array_equal(v[8:10].numpy(), vals[8:10])


def test_local_shard_view(store):
    arr = np.arange(12, dtype=np.float32).reshape(6, 2)
    store.add("x", arr)
    sh = store.local_shard("x")
    assert sh.shape == (6, 2)
    assert np.array_equal(sh.numpy(), arr)


def test_reshuffle_single_rank(store):
    from ddstore_amd.reshuffle import expected_perm

    arr = np.arange(64, dtype=np.float32).reshape(16, 4)
    store.add("x", arr)
    store.reshuffle("x", seed=123)
    perm = expected_perm(16, 123, store.device).numpy()
    out = store.get_batch("x", list(range(16)))
    assert np.array_equal(out.numpy(), arr[perm])


def test_stats(store):
    store.add("x", np.zeros((8, 4), dtype=np.float32))
    store.get_batch("x", [0, 1, 2])
    st = store.stats()["x"]
    assert st["n_gather"] == 1 and st["rows_gathered"] == 3
    assert st["bytes_gathered"] == 3 * 4 * 4


def test_method_param_compat():
    # both reference transports map onto the single native path
    s0 = DDStore(device="cpu", method=0)
    s1 = DDStore(device="cpu", method=1)
    for s in (s0, s1):
        s.add("x", np.ones((4, 2), dtype=np.float32))
        out = np.zeros((2, 2), dtype=np.float32)
        s.get("x", out, 1)
        assert (out == 1).all()
        s.free()


def test_free_var_and_readd(store):
    store.add("tmp", np.ones((4, 2), dtype=np.float32))
    store._backend.free_var("tmp")
    del store._vars["tmp"]
    store.add("tmp", 2 * np.ones((4, 2), dtype=np.float32))
    out = store.get_batch("tmp", [0])
    assert out[0, 0].item() == 2.0


def test_free_idempotent():
    from ddstore_amd import DDStore

    s = DDStore(device="cpu")
    s.add("x", np.zeros((2, 2), dtype=np.float32))
    s.free()
    s.free()  # must not raise


def test_two_stores_coexist():
    from ddstore_amd import DDStore

    a = DDStore(device="cpu")
    b = DDStore(device="cpu")
    a.add("x", np.ones((4, 2), dtype=np.float32))
    b.add("x", 5 * np.ones((4, 2), dtype=np.float32))
    assert a.get_batch("x", [0])[0, 0].item() == 1.0
    assert b.get_batch("x", [0])[0, 0].item() == 5.0
    a.free()
    b.free()


def test_many_variables(store):
    for i in range(8):
        store.add(f"m{i}", np.full((4, 2), float(i), dtype=np.float32))
    for i in range(8):
        assert store.get_batch(f"m{i}", [3])[0, 0].item() == float(i)


def test_gather_index_out_of_range(store):
    store.add("x", np.zeros((5, 2), dtype=np.float32))
    with pytest.raises(RuntimeError, match="out of range"):
        store.get_batch("x", [5])


def test_dump_load_roundtrip(tmp_path, store):
    arr = np.random.rand(16, 4).astype(np.float32)
    store.add("ck", arr)
    store.dump("ck", str(tmp_path / "ck.pt"))
    s2 = DDStore(device="cpu")
    s2.load("ck", str(tmp_path / "ck.pt"))
    out = s2.get_batch("ck", list(range(16)))
    assert np.array_equal(out.numpy(), arr)
    s2.free()


def test_dump_load_csr(tmp_path, store):
    lengths = [3, 1, 4]
    vals = np.random.rand(8, 2).astype(np.float64)
    store.add_csr("ckc", vals, lengths)
    store.dump("ckc", str(tmp_path / "ckc.pt"))
    s2 = DDStore(device="cpu")
    s2.load("ckc", str(tmp_path / "ckc.pt"))
    v, off = s2.get_csr("ckc", [2, 0, 1])
    assert off.tolist() == [0, 4, 7, 8]
    assert np.array_equal(v[:4].numpy(), vals[4:8])
    s2.free()


def test_get_noncontiguous_output_rejected(store):
    store.add("x", np.zeros((8, 4), dtype=np.float32))
    bad = np.zeros((4, 8), dtype=np.float32)[:, ::2]
    with pytest.raises(ValueError, match="contiguous"):
        store.get("x", bad, 0)
    badt = torch.zeros(8, 4)[:, ::2].clone().t()
    with pytest.raises(ValueError, match="contiguous"):
        store.get("x", badt.t()[:, ::2] if not badt.is_contiguous() else badt.t(), 0)


def test_failed_add_leaves_no_entry(store):
    bad = np.zeros((4, 8), dtype=np.float32)[:, ::2]  # non-contiguous
    with pytest.raises((RuntimeError, TypeError, ValueError)):
        # bypass Python normalization to hit the native contiguity check
        store._backend.add("ghost", torch.from_numpy(np.zeros((4, 8), dtype=np.float32))[:, ::2], 4, 4, [4])
    # registration must not have happened
    with pytest.raises(RuntimeError, match="unknown variable"):
        store.query("ghost")
    # and the name is reusable
    store.add("ghost", np.ones((4, 4), dtype=np.float32))
    assert store.get_batch("ghost", [0])[0, 0].item() == 1.0


def test_reshuffle_csr_single(store):
    from ddstore_amd.reshuffle import expected_perm

    lengths = [2, 5, 1, 3]
    vals = np.concatenate(
        [np.full(l, i, dtype=np.float64) for i, l in enumerate(lengths)]
    ).reshape(-1, 1)
    store.add_csr("rc", vals, lengths)
    store.reshuffle("rc", seed=5)
    perm = expected_perm(4, 5, store.device).tolist()
    v, off = store.get_csr("rc", [0, 1, 2, 3])
    off = off.tolist()
    for j in range(4):
        seg = v[off[j] : off[j + 1], 0].numpy()
        assert len(seg) == lengths[perm[j]] and (seg == perm[j]).all()


@pytest.mark.parametrize("fp8", [torch.float8_e4m3fn, torch.float8_e5m2])
def test_fp8_store_roundtrip(store, fp8):
    arr = torch.randn(32, 8).to(fp8)
    store.add(f"f8{str(fp8)[-4:]}", arr)
    out = store.get_batch(f"f8{str(fp8)[-4:]}", [5, 0, 31])
    assert out.dtype == fp8
    assert torch.equal(out.view(torch.uint8), arr[[5, 0, 31]].view(torch.uint8))
    # CPU cast path (tmp + .to)
    outf = store.get_batch(f"f8{str(fp8)[-4:]}", [1, 2], dtype=torch.float32)
    assert torch.equal(outf, arr[[1, 2]].to(torch.float32))


def test_affine_gather_cpu(store):
    arr = np.random.randint(0, 255, (16, 4)).astype(np.uint8)
    store.add("af", arr)
    out = store.get_batch("af", [2, 9], dtype=torch.float32, affine=(2.0, -1.0))
    ref = torch.from_numpy(arr[[2, 9]]).to(torch.float32) * 2.0 - 1.0
    assert torch.allclose(out, ref)
    with pytest.raises(TypeError, match="affine output"):
        store.get_batch("af", [0], dtype=torch.int32, affine=(1.0, 0.0))


def test_prefetch_csr_cpu(store):
    from ddstore_amd import PrefetchLoader

    lengths = [3, 1, 4, 2, 5, 2, 1, 6]
    vals = np.concatenate(
        [np.full(l, i, dtype=np.float32) for i, l in enumerate(lengths)]
    ).reshape(-1, 1)
    store.add_csr("pc", vals, lengths)
    order = [7, 0, 3, 5, 1, 2, 6, 4]
    got = []
    for (v, off) in PrefetchLoader(store, "pc", order, batch_size=3):
        off = off.tolist()
        for k in range(len(off) - 1):
            got.append(v[off[k] : off[k + 1], 0].tolist())
    for k, g in enumerate(order):
        assert got[k] == [float(g)] * lengths[g], (k, g)


def test_prefetch_ragged_tail(store):
    # final batch smaller than batch_size must not reuse a full-size buffer
    from ddstore_amd import PrefetchLoader

    arr = np.arange(50, dtype=np.float32).reshape(50, 1)
    store.add("rt", arr)
    sizes = [b.shape[0] for b in PrefetchLoader(store, "rt", np.arange(50), 16)]
    assert sizes == [16, 16, 16, 2]


def test_epoch_context_manager(store):
    store.add("ec", np.ones((4, 2), dtype=np.float32))
    with store.epoch():
        out = store.get_batch("ec", [1])
        assert out[0, 0] == 1.0
    # closed: a fresh begin works
    store.epoch_begin()
    store.epoch_end()
    # exceptions still close the epoch
    with pytest.raises(RuntimeError):
        with store.epoch():
            store.get("ec", np.zeros((9, 2), dtype=np.float32), 0)  # invalid count
    store.epoch_begin()
    store.epoch_end()


def test_fuzz_shadow_model():
    import sys, os
    sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "tools"))
    from fuzz_store import run as fuzz_run

    n = fuzz_run(ops=150, seed=99, device="cpu")
    assert n > 20


def test_io_single_rank(tmp_path, store):
    from ddstore_amd import io as dio

    arr = np.random.rand(30, 5).astype(np.float64)
    np.save(tmp_path / "a.npy", arr)
    dio.add_from_npy(store, "a", str(tmp_path / "a.npy"))
    out = store.get_batch("a", list(range(30)))
    assert np.array_equal(out.numpy(), arr)
    raw = np.random.rand(12, 6).astype(np.float32)
    raw.tofile(tmp_path / "b.bin")
    dio.add_from_memmap(store, "b", str(tmp_path / "b.bin"), np.float32, (6,))
    assert store.query("b")["nrows_total"] == 12
    with pytest.raises(ValueError, match="whole number of rows"):
        dio.add_from_memmap(store, "bad", str(tmp_path / "b.bin"), np.float32, (7,))


def test_variables_listing(store):
    assert store.variables() == []
    store.add("v1", np.zeros((2, 2), dtype=np.float32))
    store.add_csr("v2", np.zeros((3, 1), dtype=np.float32), [1, 2])
    assert sorted(store.variables()) == ["v1", "v2"]


def test_dump_load_after_reshuffle(tmp_path, store):
    from ddstore_amd.reshuffle import expected_perm

    arr = np.arange(40, dtype=np.float32).reshape(20, 2)
    store.add("dr", arr)
    store.reshuffle("dr", seed=3)
    store.dump("dr", str(tmp_path / "dr.pt"))
    s2 = DDStore(device="cpu")
    s2.load("dr", str(tmp_path / "dr.pt"))
    perm = expected_perm(20, 3, store.device).numpy()
    out = s2.get_batch("dr", list(range(20)))
    assert np.array_equal(out.numpy(), arr[perm])
    s2.free()


def test_reshuffle_inside_epoch_rejected(store):
    store.add("re", np.zeros((8, 4), dtype=np.float32))
    store.epoch_begin()
    with pytest.raises(RuntimeError, match="open epoch"):
        store.reshuffle("re", seed=1)
    store.epoch_end()
    store.reshuffle("re", seed=1)  # fine outside


def test_device_arg_forms():
    # int / "cpu" device argument forms
    s = DDStore(device="cpu")
    assert s.mode == "shm"
    s.free()


def test_init_explicit_dtype(store):
    store.init("ex", 6, 3, dtype=torch.int64)
    store.update("ex", np.arange(6, dtype=np.int64).reshape(2, 3), offset=1)
    out = store.get_batch("ex", [0, 1, 2])
    assert out.dtype == torch.int64
    assert out[0].sum() == 0 and out[1].tolist() == [0, 1, 2]


def test_prefetch_len_and_drop_last(store):
    from ddstore_amd import PrefetchLoader

    store.add("pl", np.zeros((50, 2), dtype=np.float32))
    keep = PrefetchLoader(store, "pl", np.arange(50), 16)
    drop = PrefetchLoader(store, "pl", np.arange(50), 16, drop_last=True)
    assert len(keep) == 4 and len(drop) == 3
    assert sum(b.shape[0] for b in keep) == 50
    assert sum(b.shape[0] for b in drop) == 48
