"""Unit tests for the communication layer and small helpers."""
import numpy as np
import pytest
import torch
import torch.distributed as dist

from ddstore_amd import Comm, as_comm, nsplit
from tests.dist_utils import run_dist

pytestmark = pytest.mark.timeout(300)


def test_self_comm():
    c = Comm()
    assert c.rank == 0 and c.size == 1
    assert c.Get_rank() == 0 and c.Get_size() == 1
    assert c.allgather(42) == [42]
    assert c.bcast("x") == "x"
    assert c.allreduce_max_int(7) == 7
    c.barrier()  # no-op
    assert c.Split(0, 0) is c


def test_as_comm_identity_and_wrap():
    c = Comm()
    assert as_comm(c) is c
    assert as_comm(None).size == 1

    class FakeMpi:
        def Get_rank(self):
            return 0

        def Get_size(self):
            return 1

        def allgather(self, obj):
            return [obj]

        def Barrier(self):
            pass

        def bcast(self, obj, root=0):
            return obj

    w = as_comm(FakeMpi())
    assert w.rank == 0 and w.allgather(5) == [5]


def test_self_all_to_all():
    c = Comm()
    src = torch.arange(12).reshape(4, 3)
    dst = torch.empty_like(src)
    c.all_to_all_single(dst, src, [4], [4])
    assert torch.equal(dst, src)


def test_nsplit():
    assert nsplit(10, 3) == [4, 3, 3]
    assert nsplit(9, 3) == [3, 3, 3]
    assert nsplit(2, 4) == [1, 1, 0, 0]
    assert sum(nsplit(12345, 7)) == 12345


# --------------------------------------------------------------------------
def _w_collectives(rank, world):
    c = Comm()
    assert c.size == world and c.rank == rank
    assert c.allgather(rank) == list(range(world))
    assert c.bcast("r0" if rank == 0 else None) == "r0"
    assert c.allreduce_max_int(rank * 10) == (world - 1) * 10
    # all_to_all_v with uneven splits over gloo (pairwise fallback)
    send = torch.full((rank + 1,), float(rank))
    # rank r sends (r+1) items replicated to every peer? build per-dest splits:
    sendbuf = torch.cat([send for _ in range(world)])
    in_splits = [rank + 1] * world
    out_splits = [p + 1 for p in range(world)]
    recv = torch.empty(sum(out_splits))
    c.all_to_all_single(recv, sendbuf, out_splits, in_splits)
    off = 0
    for p in range(world):
        seg = recv[off : off + p + 1]
        assert (seg == float(p)).all()
        off += p + 1


def test_collectives_ws3():
    run_dist(_w_collectives, 3)


def _w_split_keys(rank, world):
    c = Comm()
    # order-preserving key works (the reference's own pattern, key=rank)
    g = c.Split(color=0, key=rank)
    assert g.size == world and g.rank == rank
    # a REORDERING key cannot be honored on torch.distributed -> explicit error
    try:
        c.Split(color=0, key=-rank)
        raise AssertionError("expected NotImplementedError")
    except NotImplementedError:
        pass


def test_split_key_ordering():
    run_dist(_w_split_keys, 3)


def _w_bcast_nonzero_root(rank, world):
    c = Comm()
    v = c.bcast("from2" if rank == 2 else None, root=2)
    assert v == "from2"


def test_bcast_nonzero_root():
    run_dist(_w_bcast_nonzero_root, 3)
