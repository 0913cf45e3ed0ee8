"""In-place chunked reshuffle (BASELINE config 4 at capacity): must produce
exactly the same slot-preserving result as the all-to-all reshuffle, with
O(chunk) transient memory. Exercises the native cycle_order traversal and
the cycle-head save for closing writes."""
import numpy as np
import pytest
import torch

from ddstore_amd import DDStore
from ddstore_amd import _C
from ddstore_amd.reshuffle import expected_perm, reshuffle_epoch_chunked
from tests.dist_utils import run_dist


def test_cycle_order_properties():
    for seed in range(5):
        n = [1, 2, 17, 256, 1000][seed]
        perm = torch.randperm(n, generator=torch.Generator().manual_seed(seed))
        order, starts = _C.cycle_order(perm)
        assert order.shape == (n,)
        assert sorted(order.tolist()) == list(range(n))  # a traversal of all slots
        s = starts.tolist()
        assert s[0] == 0 and s[-1] == n
        p = perm.tolist()
        o = order.tolist()
        for c in range(len(s) - 1):
            for i in range(s[c], s[c + 1] - 1):
                assert p[o[i]] == o[i + 1]  # within-cycle successor = perm
            assert p[o[s[c + 1] - 1]] == o[s[c]]  # cycle closes on its head


def test_cycle_order_rejects_non_permutation():
    with pytest.raises(RuntimeError, match="not a permutation"):
        _C.cycle_order(torch.tensor([0, 0, 2], dtype=torch.int64))
    with pytest.raises(RuntimeError, match="not a permutation"):
        _C.cycle_order(torch.tensor([1, 2, 3], dtype=torch.int64))


@pytest.mark.parametrize("chunk_bytes", [64, 1024, 1 << 20])
def test_chunked_matches_expected_single(chunk_bytes):
    s = DDStore(device="cpu")
    n, d = 257, 8
    arr = torch.arange(n, dtype=torch.float32).unsqueeze(1).repeat(1, d)
    s.add("x", arr)
    reshuffle_epoch_chunked(s, "x", seed=21, max_chunk_bytes=chunk_bytes)
    perm = expected_perm(n, 21, "cpu")
    out = s.get_batch("x", list(range(n)))
    assert torch.equal(out, arr[perm])
    s.free()


def test_chunked_equals_alltoall_result():
    a = DDStore(device="cpu")
    b = DDStore(device="cpu")
    arr = torch.randn(300, 4)
    a.add("x", arr)
    b.add("x", arr.clone())
    a.reshuffle("x", seed=77)  # all-to-all path
    reshuffle_epoch_chunked(b, "x", seed=77, max_chunk_bytes=512)
    ga = a.get_batch("x", list(range(300)))
    gb = b.get_batch("x", list(range(300)))
    assert torch.equal(ga, gb)
    a.free()
    b.free()


def _w_chunked_ws2(rank, world):
    s = DDStore(device="cpu")
    n = 128
    base = torch.arange(rank * n, (rank + 1) * n, dtype=torch.float32)
    s.add("x", base.unsqueeze(1).repeat(1, 4))
    reshuffle_epoch_chunked(s, "x", seed=13, max_chunk_bytes=256)
    perm = expected_perm(n * world, 13, "cpu")
    out = s.get_batch("x", list(range(n * world)))
    assert torch.equal(out[:, 0], perm.to(torch.float32))
    s.free()


def test_chunked_ws2():
    run_dist(_w_chunked_ws2, 2)


def _w_chunked_ws3_uneven(rank, world):
    # uneven shards: 50 / 100 / 30 rows
    sizes = [50, 100, 30]
    s = DDStore(device="cpu")
    lo = sum(sizes[:rank])
    base = torch.arange(lo, lo + sizes[rank], dtype=torch.float32)
    s.add("x", base.unsqueeze(1).repeat(1, 2))
    reshuffle_epoch_chunked(s, "x", seed=99, max_chunk_bytes=128)
    n = sum(sizes)
    perm = expected_perm(n, 99, "cpu")
    out = s.get_batch("x", list(range(n)))
    assert torch.equal(out[:, 0], perm.to(torch.float32))
    s.free()


def test_chunked_ws3_uneven():
    run_dist(_w_chunked_ws3_uneven, 3)


def test_chunked_rejects_open_epoch():
    s = DDStore(device="cpu")
    s.add("x", torch.zeros(8, 2))
    s.epoch_begin()
    with pytest.raises(RuntimeError, match="open epoch"):
        reshuffle_epoch_chunked(s, "x", 1)
    s.epoch_end()
    s.free()


def test_chunked_rejects_csr():
    s = DDStore(device="cpu")
    s.add_csr("c", torch.zeros(6, 1), [2, 4])
    with pytest.raises(ValueError, match="fixed-stride"):
        reshuffle_epoch_chunked(s, "c", 1)
    s.free()


@pytest.mark.parametrize("n", [1, 2, 5, 64, 257, 5000])
def test_cycle_order_device_matches_native(n):
    """The pointer-doubling traversal must equal the serial walk EXACTLY
    (same cycle order, same within-cycle order) -- torch ops run this on
    CPU here; on GPU the same code path feeds the capacity reshuffle."""
    from ddstore_amd.reshuffle import cycle_order_device

    for seed in range(3):
        perm = torch.randperm(n, generator=torch.Generator().manual_seed(seed))
        o1, s1 = _C.cycle_order(perm)
        o2, s2 = cycle_order_device(perm)
        assert torch.equal(o1, o2.cpu()), (n, seed)
        assert torch.equal(s1, s2.cpu()), (n, seed)
    # identity and reversal corner cases
    for perm in [torch.arange(n), torch.arange(n - 1, -1, -1)]:
        o1, s1 = _C.cycle_order(perm.contiguous())
        o2, s2 = cycle_order_device(perm.contiguous())
        assert torch.equal(o1, o2.cpu()) and torch.equal(s1, s2.cpu())


def test_cycle_order_device_property():
    """Hypothesis-style randomized check across many shapes: the parallel
    ranking equals the serial walk for every permutation tried."""
    from ddstore_amd.reshuffle import cycle_order_device

    g = torch.Generator().manual_seed(2024)
    for _ in range(60):
        n = int(torch.randint(1, 3000, (1,), generator=g))
        perm = torch.randperm(n, generator=g)
        o1, s1 = _C.cycle_order(perm)
        o2, s2 = cycle_order_device(perm)
        assert torch.equal(o1, o2) and torch.equal(s1, s2), n
