"""Round-2 fix coverage: CSR init/update, capacity clamping,
host CSR bounds checks, strict-OOB mode, CPU gather dtype
agreement, Split group caching, reshuffle permutation-agreement check.
"""
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

from ddstore_amd import DDStore
from tests.dist_utils import run_dist

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture
def store():
    s = DDStore(device="cpu")
    yield s
    s.free()


# ---------------------------------------------------------------- CSR init/update
def test_csr_init_update_roundtrip(store):
    lens = torch.tensor([3, 1, 4, 2])
    store.init_csr("c", lens, disp=2, dtype=torch.float32)
    # zeroed until filled
    v, off = store.get_csr("c", [0, 1, 2, 3])
    assert v.abs().sum() == 0 and off.tolist() == [0, 3, 4, 8, 10]
    # fill samples [0,2) then [2,4) separately (incremental-fill pattern)
    part0 = torch.arange(8, dtype=torch.float32).reshape(4, 2)  # 3+1 elems
    part1 = 100 + torch.arange(12, dtype=torch.float32).reshape(6, 2)  # 4+2
    store.update_csr("c", part0, offset=0)
    store.update_csr("c", part1, offset=2)
    v, off = store.get_csr("c", [0, 1, 2, 3])
    expect = torch.cat([part0, part1])
    assert torch.equal(v, expect)


def test_csr_update_out_of_range(store):
    store.init_csr("c", [2, 2], disp=1)
    with pytest.raises(Exception, match="out of range"):
        store.update_csr("c", torch.zeros(5, 1), offset=0)
    with pytest.raises(IndexError):
        store.update_csr("c", torch.zeros(1, 1), offset=7)


def test_csr_update_non_csr_rejected(store):
    store.add("fx", np.zeros((4, 2), dtype=np.float32))
    with pytest.raises(ValueError, match="not a CSR variable"):
        store.update_csr("fx", torch.zeros(2, 2))


def _w_csr_init_update_ws2(rank, world):
    s = DDStore(device="cpu")
    lens = [2 + rank, 3]  # rank 0: [2,3], rank 1: [3,3]
    s.init_csr("c", lens, disp=1, dtype=torch.float64)
    vals = torch.full((sum(lens), 1), float(rank + 1), dtype=torch.float64)
    s.update_csr("c", vals, offset=0)
    s.comm.barrier()
    v, off = s.get_csr("c", list(range(2 * world)))
    off = off.tolist()
    # sample i belongs to rank i//2 -> value rank+1
    for i in range(2 * world):
        seg = v[off[i] : off[i + 1]]
        assert (seg == float(i // 2 + 1)).all()
    s.free()


def test_csr_init_update_ws2():
    run_dist(_w_csr_init_update_ws2, 2)


# ------------------------------------------------------------- capacity clamping
def test_csr_capacity_clamp_host(store):
    lens = torch.tensor([4, 4, 4, 4])
    vals = torch.arange(16, dtype=torch.float32).reshape(16, 1)
    store.add_csr("c", vals, lens)
    # capacity for only 2 of the 4 requested samples: the overflowing ones
    # are SKIPPED and counted, memory is never written past the buffer
    out = torch.zeros(8, 1, dtype=torch.float32)
    v, off = store.get_csr("c", [0, 1, 2, 3], out=out)
    q = store.query("c")
    assert q["cap_skipped"] == 2
    assert torch.equal(v[0:4], vals[0:4]) and torch.equal(v[4:8], vals[4:8])


def test_csr_oob_index_host_skipped(store):
    lens = torch.tensor([2, 2])
    vals = torch.arange(4, dtype=torch.float32).reshape(4, 1)
    store.add_csr("c", vals, lens)
    out = torch.full((4, 1), -1.0)
    off = torch.tensor([0, 2, 4])  # sample 99 would land at [2,4)
    store._backend.gather_csr("c", torch.tensor([0, 99]), off, out, 4)
    q = store.query("c")
    assert q["oob_skipped"] == 1
    assert torch.equal(out[0:2], vals[0:2])
    assert (out[2:4] == -1.0).all()  # untouched, not garbage


def test_csr_stats_true_bytes(store):
    lens = torch.tensor([3, 1])
    vals = torch.arange(4, dtype=torch.float32).reshape(4, 1)
    store.add_csr("c", vals, lens)
    big = torch.zeros(64, 1, dtype=torch.float32)  # oversized capacity buffer
    store.get_csr("c", [0, 1], out=big)
    q = store.query("c")
    assert q["bytes_gathered"] == 4 * 4  # true gathered bytes, not capacity


# ------------------------------------------------------------------ strict mode
_STRICT_SNIPPET = r"""
import torch
from ddstore_amd import DDStore
s = DDStore(device="cpu")
s.add_csr("c", torch.arange(4, dtype=torch.float32).reshape(4, 1), [2, 2])
out = torch.zeros(4, 1)
off = torch.tensor([0, 2, 4])
s.epoch_begin()
s._backend.gather_csr("c", torch.tensor([0, 99]), off, out, 4)
try:
    s.epoch_end()
    print("NO-RAISE")
except RuntimeError as e:
    assert "DDSTORE_STRICT" in str(e), e
    print("RAISED-OK")
s._backend.free_all()
"""


def _run_snippet(code: str, **env):
    e = dict(os.environ, **{k: str(v) for k, v in env.items()})
    return subprocess.run(
        [sys.executable, "-c", code], cwd=REPO, env=e,
        capture_output=True, text=True, timeout=120,
    )


def test_strict_mode_raises_subprocess():
    r = _run_snippet(_STRICT_SNIPPET, DDSTORE_STRICT="1")
    assert r.returncode == 0, r.stderr
    assert "RAISED-OK" in r.stdout


def test_default_mode_does_not_raise_subprocess():
    r = _run_snippet(_STRICT_SNIPPET, DDSTORE_STRICT="0")
    assert r.returncode == 0, r.stderr
    assert "NO-RAISE" in r.stdout


def test_reset_counters(store):
    store.add("x", np.ones((4, 4), dtype=np.float32))
    tmp = torch.empty(2, 4, dtype=torch.float32)
    store._backend.gather("x", torch.tensor([0, 1]), tmp)
    assert store.query("x")["rows_gathered"] == 2
    store.reset_counters("x")
    q = store.query("x")
    assert q["rows_gathered"] == 0 and q["oob_skipped"] == 0


# ------------------------------------------------------- CPU gather dtype check
def test_cpu_gather_same_size_dtype_rejected(store):
    # f16 store, bf16 out: same itemsize -- a byte move would silently
    # reinterpret bits; the native host gather now rejects, and
    # get_batch converts via staging instead
    store.add("h", torch.ones(4, 4, dtype=torch.float16))
    out = torch.empty(2, 4, dtype=torch.bfloat16)
    with pytest.raises(RuntimeError, match="store dtype"):
        store._backend.gather("h", torch.tensor([0, 1]), out)
    got = store.get_batch("h", [0, 1], dtype=torch.bfloat16)
    assert got.dtype == torch.bfloat16 and (got == 1.0).all()


# ------------------------------------------------------------- Split group cache
def _w_split_cached(rank, world):
    from ddstore_amd.comm import _group_cache

    a = DDStore(device="cpu", ddstore_width=2)
    n0 = len(_group_cache)
    assert n0 >= 1
    b = DDStore(device="cpu", ddstore_width=2)
    assert len(_group_cache) == n0  # second Split reused the cached groups
    a.add("x", np.full((4, 2), float(a.rank + 1), dtype=np.float32))
    b.add("x", np.full((4, 2), float(b.rank + 10), dtype=np.float32))
    assert a.get_batch("x", [5])[0, 0] == 2.0
    assert b.get_batch("x", [1])[0, 0] == 10.0
    a.free()
    b.free()


def test_split_group_cache_ws4():
    run_dist(_w_split_cached, 4)


# ------------------------------------------------- reshuffle perm agreement
def _w_perm_divergence(rank, world):
    from ddstore_amd.reshuffle import _check_perm_agreement

    s = DDStore(device="cpu")
    good = torch.randperm(64, generator=torch.Generator().manual_seed(5))
    _check_perm_agreement(s, good)  # identical everywhere: fine
    bad = torch.randperm(64, generator=torch.Generator().manual_seed(rank))
    try:
        _check_perm_agreement(s, bad)
        raise AssertionError("divergent perms not detected")
    except RuntimeError as e:
        assert "DIFFERENT permutations" in str(e)
    s.free()


def test_perm_divergence_detected_ws2():
    run_dist(_w_perm_divergence, 2)


def test_concurrent_gathers_thread_safe(store):
    """The heavy native ops release the GIL (r2), so several Python threads
    can gather at once; HostPool serializes its task state internally.
    Verifies results stay correct under 4 concurrent gather threads."""
    import threading

    rows, dim = 20000, 16
    base = torch.arange(rows, dtype=torch.float32).unsqueeze(1).repeat(1, dim)
    store.add("mt", base)
    errs = []

    def worker(seed):
        try:
            rng = np.random.default_rng(seed)
            for _ in range(20):
                idx = torch.from_numpy(rng.integers(0, rows, size=4096))
                out = store.get_batch("mt", idx)
                assert torch.equal(out[:, 0], idx.to(torch.float32))
        except Exception as e:  # pragma: no cover
            errs.append(e)

    ts = [threading.Thread(target=worker, args=(s,)) for s in range(4)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert not errs, errs
