"""Multi-process (gloo, CPU) store tests -- the distributed semantics the
reference exercises under ``mpirun`` (test/demo.py, test/test.py),
run here with spawned torch.distributed ranks on 127.0.0.1.

Self-verifying data pattern: rank r fills its shard with the constant r+1 so
a fetched row proves who owned it (reference test/demo.py:35-39)."""
import numpy as np
import pytest
import torch
import torch.distributed as dist

from tests.dist_utils import run_dist

pytestmark = pytest.mark.timeout(300)

NUM, DIM = 256, 16


def _mkstore(rank, world, width=None):
    from ddstore_amd import DDStore

    return DDStore(device="cpu", ddstore_width=width)


# --------------------------------------------------------------------------
def _w_demo(rank, world):
    """demo.py parity: add -> epoched random gets -> value check -> free."""
    from ddstore_amd import DDStore

    s = DDStore(device="cpu")
    arr = np.full((NUM, DIM), rank + 1, dtype=np.float64)
    s.add("x", arr)
    rng = np.random.default_rng(1000 + rank)
    for _ in range(8):
        s.epoch_begin()
        idx = int(rng.integers(0, NUM * world))
        out = np.zeros((1, DIM), dtype=np.float64)
        s.get("x", out, start=idx)
        assert out.mean() == idx // NUM + 1, (idx, out.mean())
        s.epoch_end()
    s.free()


def test_demo_parity_ws2():
    run_dist(_w_demo, 2)


def test_demo_parity_ws4():
    run_dist(_w_demo, 4)


# --------------------------------------------------------------------------
def _w_batch(rank, world):
    from ddstore_amd import DDStore

    s = DDStore(device="cpu")
    arr = np.full((NUM, DIM), rank + 1, dtype=np.float32)
    s.add("x", arr)
    rng = np.random.default_rng(7)  # same on all ranks
    idx = rng.integers(0, NUM * world, size=64)
    out = s.get_batch("x", idx)
    expect = (idx // NUM + 1).astype(np.float32)
    assert np.array_equal(out.numpy()[:, 0], expect)
    # global shuffle sum invariant
    total = out.numpy().sum()
    expected = expect.sum() * DIM
    assert total == expected
    s.free()


def test_get_batch_global(ws=2):
    run_dist(_w_batch, ws)


# --------------------------------------------------------------------------
def _w_boundary(rank, world):
    from ddstore_amd import DDStore

    s = DDStore(device="cpu")
    s.add("x", np.zeros((NUM, DIM), dtype=np.float32))
    out = np.zeros((2, DIM), dtype=np.float32)
    # crossing the shard boundary must throw (reference ddstore.hpp:210-214)
    try:
        s.get("x", out, start=NUM - 1)
        raise AssertionError("expected Invalid count on target")
    except RuntimeError as e:
        assert "Invalid count on target" in str(e)
    s.free()


def test_cross_shard_get_raises():
    run_dist(_w_boundary, 2)


# --------------------------------------------------------------------------
def _w_csr(rank, world):
    from ddstore_amd import DDStore

    s = DDStore(device="cpu")
    # rank r owns samples with lengths [r+1, r+2], values == global sample id
    lengths = [rank + 1, rank + 2]
    nelem = sum(lengths)
    gid0 = 2 * rank
    vals = np.concatenate(
        [np.full(l, gid0 + i, dtype=np.float32) for i, l in enumerate(lengths)]
    ).reshape(-1, 1)
    s.add_csr("c", vals, lengths)
    # every rank reads every sample
    ntotal = 2 * world
    v, off = s.get_csr("c", list(range(ntotal)))
    off = off.tolist()
    for g in range(ntotal):
        seg = v[off[g] : off[g + 1], 0].numpy()
        expected_len = (g // 2) + 1 + (g % 2)
        assert len(seg) == expected_len, (g, len(seg), expected_len)
        assert (seg == g).all()
    s.free()


def test_csr_remote():
    run_dist(_w_csr, 3)


# --------------------------------------------------------------------------
def _w_reshuffle(rank, world):
    from ddstore_amd import DDStore

    s = DDStore(device="cpu")
    base = np.arange(rank * NUM, (rank + 1) * NUM, dtype=np.float32)
    arr = np.repeat(base[:, None], DIM, axis=1)  # row g == value g
    s.add("x", arr)
    s.reshuffle("x", seed=99)
    from ddstore_amd.reshuffle import expected_perm

    perm = expected_perm(NUM * world, 99, s.device).numpy()
    out = s.get_batch("x", list(range(NUM * world)))
    assert np.array_equal(out.numpy()[:, 0], perm.astype(np.float32))
    s.free()


def test_reshuffle_ws2():
    run_dist(_w_reshuffle, 2)


def test_reshuffle_ws4():
    run_dist(_w_reshuffle, 4)


# --------------------------------------------------------------------------
def _w_coexist(rank, world):
    """Store traffic interleaved with a DDP-style all_reduce per batch
    (reference test/test.py:153-154 coexistence probe)."""
    from ddstore_amd import DDStore

    s = DDStore(device="cpu")
    arr = np.full((NUM, DIM), rank + 1, dtype=np.float32)
    s.add("a", arr)
    s.add("b", 10 * arr)
    rng = np.random.default_rng(5)
    for _ in range(4):
        s.epoch_begin()
        idx = rng.integers(0, NUM * world, size=8)
        oa = s.get_batch("a", idx)
        ob = s.get_batch("b", idx)
        assert np.array_equal(ob.numpy(), 10 * oa.numpy())
        s.epoch_end()
        t = torch.tensor([float(rank)])
        dist.all_reduce(t)
        assert t.item() == world * (world - 1) / 2
    s.free()


def test_store_ddp_coexistence():
    run_dist(_w_coexist, 2)


# --------------------------------------------------------------------------
def _w_width(rank, world):
    """Replication groups: width=2 on 4 ranks -> two groups, each holding a
    full replica partitioned internally (reference README.md:154-172)."""
    from ddstore_amd import DDStore

    s = DDStore(device="cpu", ddstore_width=2)
    assert s.size == 2 and s.rank == rank % 2
    arr = np.full((NUM, DIM), s.rank + 1, dtype=np.float32)
    s.add("x", arr)
    q = s.query("x")
    assert q["nrows_total"] == 2 * NUM
    out = np.zeros((1, DIM), dtype=np.float32)
    s.get("x", out, start=NUM + 3)  # owned by group-rank 1
    assert (out == 2).all()
    s.free()


def test_ddstore_width_groups():
    run_dist(_w_width, 4)


# --------------------------------------------------------------------------
def _w_uniform_disp(rank, world):
    from ddstore_amd import DDStore

    s = DDStore(device="cpu")
    dim = 4 if rank == 0 else 8
    try:
        s.add("x", np.zeros((4, dim), dtype=np.float32))
        raise AssertionError("expected disp validation error")
    except ValueError as e:
        assert "uniform" in str(e)


def test_disp_uniform_validation():
    run_dist(_w_uniform_disp, 2)


# --------------------------------------------------------------------------
def _w_distdataset(rank, world):
    from ddstore_amd import DistDataset

    n = 40
    data = np.arange(n * 6, dtype=np.float32).reshape(n, 2, 3)
    labels = np.arange(n, dtype=np.int64)
    ds = DistDataset(data, labels, device="cpu")
    assert len(ds) == n
    for g in [0, n - 1, n // 2 + 1]:
        x, y = ds[g]
        assert x.shape == (2, 3)
        assert np.array_equal(x.numpy(), data[g])
        assert y.item() == g
    # prefetch-loader path with a shuffled global order
    order = np.random.default_rng(3).permutation(n)
    got = []
    for xb, yb in ds.loader(order, batch_size=7):
        assert xb.shape[1] == 6
        got.extend(yb.view(-1).tolist())
    assert got == order.tolist()
    ds.free()


def test_distdataset_ws2():
    run_dist(_w_distdataset, 2)


# --------------------------------------------------------------------------
def _w_empty_shard(rank, world):
    """A rank with ZERO rows: directory must still route reads correctly
    (zero-width prefix entries)."""
    from ddstore_amd import DDStore

    s = DDStore(device="cpu")
    n = 0 if rank == 1 else 8
    arr = np.full((n, 4), rank + 1.0, dtype=np.float32)
    s.add("x", arr)
    q = s.query("x")
    assert q["nrows_total"] == 8 * (world - 1)
    out = s.get_batch("x", list(range(q["nrows_total"])))
    expect = np.repeat([r + 1.0 for r in range(world) if r != 1], 8)
    assert np.array_equal(out.numpy()[:, 0], expect.astype(np.float32))
    s.free()


def test_empty_shard_rank():
    run_dist(_w_empty_shard, 3)


# --------------------------------------------------------------------------
def _w_reshuffle_width(rank, world):
    """Reshuffle within replication groups: each width-2 group permutes its
    own replica independently (split-comm all-to-all)."""
    from ddstore_amd import DDStore
    from ddstore_amd.reshuffle import expected_perm

    s = DDStore(device="cpu", ddstore_width=2)
    base = np.arange(s.rank * NUM, (s.rank + 1) * NUM, dtype=np.float32)
    s.add("x", np.repeat(base[:, None], DIM, axis=1))
    s.reshuffle("x", seed=7)
    perm = expected_perm(2 * NUM, 7, s.device).numpy()
    out = s.get_batch("x", list(range(2 * NUM)))
    assert np.array_equal(out.numpy()[:, 0], perm.astype(np.float32))
    s.free()


def test_reshuffle_with_width_groups():
    run_dist(_w_reshuffle_width, 4)


# --------------------------------------------------------------------------
def _w_reshuffle_csr(rank, world):
    from ddstore_amd import DDStore
    from ddstore_amd.reshuffle import expected_perm

    s = DDStore(device="cpu")
    nloc = 20
    gid0 = rank * nloc
    rng = np.random.default_rng(rank)
    lengths = rng.integers(1, 9, size=nloc)
    vals = np.concatenate(
        [np.full(l, gid0 + i, dtype=np.float32) for i, l in enumerate(lengths)]
    ).reshape(-1, 1)
    s.add_csr("c", vals, lengths)
    s.reshuffle("c", seed=21)
    ntotal = nloc * world
    perm = expected_perm(ntotal, 21, s.device).numpy()
    v, off = s.get_csr("c", list(range(ntotal)))
    off = off.tolist()
    # after reshuffle, slot j must contain OLD sample perm[j]'s payload
    all_lens = []
    for r in range(world):
        rr = np.random.default_rng(r)
        all_lens.append(rr.integers(1, 9, size=nloc))
    all_lens = np.concatenate(all_lens)
    for j in range(ntotal):
        seg = v[off[j] : off[j + 1], 0].numpy()
        src = perm[j]
        assert len(seg) == all_lens[src], (j, src)
        assert (seg == src).all(), (j, src, seg[:3])
    s.free()


def test_reshuffle_csr_ws3():
    run_dist(_w_reshuffle_csr, 3)


# --------------------------------------------------------------------------
def _w_free_after_destroy(rank, world):
    """free() after torch.distributed teardown must not raise (reference
    guards MPI_Finalized the same way, ddstore.cxx:81-83)."""
    from ddstore_amd import DDStore

    s = DDStore(device="cpu")
    s.add("x", np.ones((8, 2), dtype=np.float32))
    dist.barrier()
    dist.destroy_process_group()
    s.free()  # barrier inside must be skipped gracefully


def test_free_after_dist_destroy():
    run_dist(_w_free_after_destroy, 2)


# --------------------------------------------------------------------------
def _w_ingest_files(rank, world, tmpdir):
    from ddstore_amd import DDStore
    from ddstore_amd import io as dio

    s = DDStore(device="cpu")
    dio.add_from_npy(s, "a", f"{tmpdir}/a.npy")
    full = np.load(f"{tmpdir}/a.npy")
    out = s.get_batch("a", list(range(full.shape[0])))
    assert np.array_equal(out.numpy(), full.reshape(full.shape[0], -1))
    dio.add_from_memmap(s, "b", f"{tmpdir}/b.bin", np.float32, (6,))
    fullb = np.fromfile(f"{tmpdir}/b.bin", dtype=np.float32).reshape(-1, 6)
    outb = s.get_batch("b", list(range(fullb.shape[0])))
    assert np.array_equal(outb.numpy(), fullb)
    dio.add_csr_from_npy(s, "c", f"{tmpdir}/cv.npy", f"{tmpdir}/cl.npy")
    lens = np.load(f"{tmpdir}/cl.npy")
    vals = np.load(f"{tmpdir}/cv.npy")
    v, off = s.get_csr("c", list(range(len(lens))))
    assert np.array_equal(v.view(-1).numpy(), vals.reshape(-1))
    s.free()


def test_ingest_from_files(tmp_path):
    rng = np.random.default_rng(0)
    np.save(tmp_path / "a.npy", rng.random((50, 2, 3)).astype(np.float32))
    rng.random((40, 6)).astype(np.float32).tofile(tmp_path / "b.bin")
    lens = rng.integers(1, 7, size=30)
    np.save(tmp_path / "cl.npy", lens)
    np.save(tmp_path / "cv.npy", rng.random((int(lens.sum()), 2)).astype(np.float64))
    run_dist(_w_ingest_files, 2, str(tmp_path))


# --------------------------------------------------------------------------
def _w_verify_transport(rank, world):
    from ddstore_amd import DDStore
    from ddstore_amd.debug import verify_transport

    s = DDStore(device="cpu")
    arr = np.random.default_rng(rank).random((100, 8)).astype(np.float32)
    s.add("x", arr)
    r = verify_transport(s, "x", chunk_rows=33)
    assert r["ok"], r
    # corrupt the local shard AFTER digests would have matched -> re-verify
    s.local_shard("x")[0, 0] += 1.0
    r2 = verify_transport(s, "x")
    # every reader (incl. self) should now flag this rank as mismatched...
    # NB: owner digest is recomputed fresh, so corruption moves digest AND
    # data together -> still consistent. Instead corrupt what readers see by
    # checking digests differ from the first run:
    assert r2["ok"]  # internally consistent again
    assert r2["digests"][rank] != r["digests"][rank]
    s.free()


def test_verify_transport_ws3():
    run_dist(_w_verify_transport, 3)
