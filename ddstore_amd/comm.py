"""Communication plane for ddstore_amd.

The reference bootstraps everything over MPI collectives (Allgather/Allreduce/
Barrier/Comm_split -- SURVEY §2.4 C3/C4/C14/C15).  The MI355X-native rebuild
runs one process per GPU and uses ``torch.distributed`` -- RCCL over xGMI on
GPU nodes (backend "nccl" IS RCCL on ROCm), gloo on CPU -- for the metadata
plane, with a self-comm fallback for world_size==1 (no init required).

``Comm`` also duck-types the small mpi4py surface the reference's Python layer
uses (Get_rank/Get_size/Barrier/Split/allgather/bcast, reference
examples/vae/distdataset.py:25-30), so reference-style callers work unchanged;
a real mpi4py communicator is likewise accepted and wrapped.
"""
from __future__ import annotations

import os
from typing import Any, List, Optional

import torch
import torch.distributed as dist


def _dist_ready() -> bool:
    return dist.is_available() and dist.is_initialized()


# Process-lifetime cache of created groups keyed by their member ranks:
# torch.distributed groups are never garbage-collected implicitly, so a
# repeated Split (e.g. many DDStore(ddstore_width=...) constructions in one
# process) must reuse instead of leaking a new group per call.
_group_cache: dict = {}


class Comm:
    """Collective metadata plane over torch.distributed (or a no-op self comm)."""

    def __init__(self, group: Optional[object] = None):
        if _dist_ready():
            self._group = group  # None => WORLD
            self._rank = dist.get_rank(group) if group is not None else dist.get_rank()
            self._size = (
                dist.get_world_size(group) if group is not None else dist.get_world_size()
            )
            # global ranks of this group's members, in group order
            if group is None:
                self._global_ranks = list(range(dist.get_world_size()))
            else:
                self._global_ranks = dist.get_process_group_ranks(group)
        else:
            self._group = None
            self._rank = 0
            self._size = 1
            self._global_ranks = [0]

    # -- basic topology ------------------------------------------------------
    @property
    def rank(self) -> int:
        return self._rank

    @property
    def size(self) -> int:
        return self._size

    @property
    def group(self):
        return self._group

    def global_rank_of(self, group_rank: int) -> int:
        return self._global_ranks[group_rank]

    # mpi4py-style aliases
    def Get_rank(self) -> int:  # noqa: N802
        return self._rank

    def Get_size(self) -> int:  # noqa: N802
        return self._size

    # -- collectives ---------------------------------------------------------
    def barrier(self) -> None:
        if self._size > 1:
            dist.barrier(group=self._group)

    def Barrier(self) -> None:  # noqa: N802
        self.barrier()

    def allgather(self, obj: Any) -> List[Any]:
        if self._size == 1:
            return [obj]
        out: List[Any] = [None] * self._size
        dist.all_gather_object(out, obj, group=self._group)
        return out

    def bcast(self, obj: Any, root: int = 0) -> Any:
        if self._size == 1:
            return obj
        box = [obj if self._rank == root else None]
        dist.broadcast_object_list(box, src=self._global_ranks[root], group=self._group)
        return box[0]

    def allreduce_max_int(self, value: int) -> int:
        return max(self.allgather(int(value)))

    def Split(self, color: int, key: int = 0) -> "Comm":  # noqa: N802
        """MPI_Comm_split semantics (reference distdataset.py:28 uses
        ``comm.Split(rank // ddstore_width, rank)`` for replication groups).

        Limitation vs MPI: torch.distributed groups always order members by
        ascending global rank, so a ``key`` that would REORDER members
        cannot be honored and raises (the reference's own usage, key=rank,
        is order-preserving)."""
        if self._size == 1:
            return self
        triples = self.allgather((int(color), int(key), self._rank))
        colors = sorted({c for c, _, _ in triples})
        my_group = None
        for c in colors:
            members = sorted(
                [(k, r) for cc, k, r in triples if cc == c]
            )  # order by (key, rank)
            if [r for _, r in members] != sorted(r for _, r in members):
                raise NotImplementedError(
                    "ddstore Comm.Split: torch.distributed orders group members "
                    "by ascending rank; a reordering `key` is not supported"
                )
            ranks = [self._global_ranks[r] for _, r in members]
            key = tuple(ranks)
            g = _group_cache.get(key)
            if g is None:
                # every rank must call new_group for every group, same order;
                # the cache is keyed identically on every rank so cache
                # hits/misses agree and the collective stays matched
                g = dist.new_group(ranks=ranks)
                _group_cache[key] = g
            if c == color:
                my_group = g
        return Comm(my_group)

    def split(self, color: int, key: int = 0) -> "Comm":
        return self.Split(color, key)

    # -- data-plane helper ---------------------------------------------------
    def all_to_all_single(
        self,
        output: torch.Tensor,
        input: torch.Tensor,
        output_split_sizes: List[int],
        input_split_sizes: List[int],
    ) -> None:
        """All-to-all-v. On RCCL this is ``dist.all_to_all_single`` over xGMI
        (the per-pair striping that reaches the 7-link aggregate, SURVEY §2.4
        C17 note); gloo lacks all_to_all so a pairwise isend/irecv schedule is
        used there."""
        if self._size == 1:
            output.copy_(input)
            return
        backend = dist.get_backend(self._group)
        if backend == "nccl":
            dist.all_to_all_single(
                output,
                input,
                output_split_sizes=output_split_sizes,
                input_split_sizes=input_split_sizes,
                group=self._group,
            )
            return
        # pairwise fallback (gloo); gloo point-to-point needs CPU tensors
        if input.is_cuda:
            out_cpu = torch.empty_like(output, device="cpu")
            self.all_to_all_single(
                out_cpu, input.cpu(), output_split_sizes, input_split_sizes
            )
            output.copy_(out_cpu)
            return
        in_off = [0]
        for s in input_split_sizes:
            in_off.append(in_off[-1] + s)
        out_off = [0]
        for s in output_split_sizes:
            out_off.append(out_off[-1] + s)
        reqs = []
        for peer in range(self._size):
            if peer == self._rank:
                continue
            chunk = input[in_off[peer] : in_off[peer + 1]]
            if chunk.numel() > 0:
                reqs.append(
                    dist.isend(
                        chunk.contiguous(),
                        dst=self._global_ranks[peer],
                        group=self._group,
                    )
                )
        recv_bufs = []
        for peer in range(self._size):
            if peer == self._rank:
                continue
            n = out_off[peer + 1] - out_off[peer]
            if n > 0:
                buf = torch.empty(
                    (n,) + tuple(input.shape[1:]), dtype=input.dtype, device=input.device
                )
                reqs.append(dist.irecv(buf, src=self._global_ranks[peer], group=self._group))
                recv_bufs.append((peer, buf))
        # local chunk
        n_self = out_off[self._rank + 1] - out_off[self._rank]
        if n_self > 0:
            output[out_off[self._rank] : out_off[self._rank + 1]] = input[
                in_off[self._rank] : in_off[self._rank + 1]
            ]
        for r in reqs:
            r.wait()
        for peer, buf in recv_bufs:
            output[out_off[peer] : out_off[peer + 1]] = buf


class _MpiWrap(Comm):
    """Adapter accepting a real mpi4py communicator (duck-typed)."""

    def __init__(self, mpi_comm):
        self._mpi = mpi_comm
        self._rank = mpi_comm.Get_rank()
        self._size = mpi_comm.Get_size()
        self._group = None
        self._global_ranks = list(range(self._size))

    def barrier(self) -> None:
        self._mpi.Barrier()

    def allgather(self, obj: Any) -> List[Any]:
        return self._mpi.allgather(obj)

    def bcast(self, obj: Any, root: int = 0) -> Any:
        return self._mpi.bcast(obj, root=root)

    def Split(self, color: int, key: int = 0) -> "Comm":  # noqa: N802
        return _MpiWrap(self._mpi.Split(color, key))

    def all_to_all_single(self, output, input, output_split_sizes, input_split_sizes):
        raise NotImplementedError(
            "reshuffle over an mpi4py comm is not supported; use torch.distributed"
        )


def as_comm(comm: Optional[object]) -> Comm:
    """Normalize a user-provided communicator: None (use torch.distributed if
    initialized, else self), a Comm, a torch.distributed ProcessGroup, or an
    mpi4py communicator."""
    if comm is None:
        return Comm()
    if isinstance(comm, Comm):
        return comm
    if hasattr(comm, "Get_rank") and hasattr(comm, "allgather"):
        return _MpiWrap(comm)
    # assume torch.distributed ProcessGroup
    return Comm(comm)


def default_device_index() -> int:
    """Device for this rank: LOCAL_RANK when launched via torchrun, else 0."""
    if "LOCAL_RANK" in os.environ:
        return int(os.environ["LOCAL_RANK"]) % max(torch.cuda.device_count(), 1)
    return 0
