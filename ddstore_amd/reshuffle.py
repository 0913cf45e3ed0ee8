"""Epoch-level global data reshuffle over xGMI.

New capability vs the reference: the reference only
shuffles *indices* (DistributedSampler) -- data never moves after ``add``.
Here the data itself is redistributed at an epoch boundary with an RCCL
all-to-all(v), which stripes traffic pairwise across all 7 xGMI links per
GPU (a ring collective would be single-link bound; pairwise all-to-all
reaches the ~1 TB/s per-GPU aggregate).

The permutation is slot-preserving: after ``reshuffle_epoch(store, name,
seed)``, global slot ``j`` holds the row that was previously at slot
``perm[j]`` -- shard sizes and the prefix directory are unchanged, only the
contents move. Every rank derives the same permutation from ``seed``
(``torch.randperm`` with a seeded generator on the store device -- identical
across ranks of one node, same device architecture), so the exchange needs
no coordinator. All plan computation (owner lookup, split sizes, placement
order) runs as torch ops on the store device; the data path is
gather-kernel -> all_to_all_single -> scatter-kernel.
"""
from __future__ import annotations

import math

import torch

from .store import DDStore


def _check_perm_agreement(store: DDStore, perm: torch.Tensor) -> None:
    """Every rank derives the permutation independently from the seed
    (device RNG); if any rank disagreed (different ROCm version/arch mixing)
    the exchange would silently corrupt data. A strided checksum allgather
    catches divergence before any data moves."""
    if store.size == 1:
        return
    n = perm.numel()
    stride = max(1, n // 4096)
    sample = perm[::stride]
    pos = torch.arange(1, sample.numel() + 1, device=sample.device, dtype=torch.int64)
    h = int(((sample + 1) * pos).sum().item())  # position-weighted, wraps ok
    if len(set(store.comm.allgather(h))) != 1:
        raise RuntimeError(
            "ddstore reshuffle: ranks derived DIFFERENT permutations from the "
            "same seed (device RNG mismatch across processes -- mixed ROCm "
            "versions or GPU architectures?); aborting before data exchange"
        )


def reshuffle_epoch(store: DDStore, name: str, seed: int) -> None:
    if store._backend.epoch_active():
        raise RuntimeError(
            "ddstore reshuffle: cannot move data inside an open epoch "
            "(concurrent gets would race); call epoch_end() first"
        )
    q = store.query(name)
    if q["is_csr"]:
        return _reshuffle_csr(store, name, seed)
    dev = store.device
    prefix = torch.tensor(q["prefix"], dtype=torch.int64, device=dev)
    ntotal = int(q["nrows_total"])
    disp = int(q["disp"])
    dtype = store._meta(name)["dtype"]
    size, rank = store.size, store.rank
    p0, p1 = int(q["prefix"][rank]), int(q["prefix"][rank + 1])

    g = torch.Generator(device=dev)
    g.manual_seed(int(seed))
    perm = torch.randperm(ntotal, generator=g, device=dev)
    _check_perm_agreement(store, perm)
    # rank that currently holds the row destined for slot j
    src_of_slot = torch.searchsorted(prefix, perm, right=True) - 1

    # --- send side: rows this rank owns, grouped by destination rank
    # (ascending j gives contiguous, ascending dest groups since owner(j) is
    # nondecreasing in j)
    j_send = (src_of_slot == rank).nonzero(as_tuple=True)[0]
    dest = torch.searchsorted(prefix, j_send, right=True) - 1
    send_counts = torch.bincount(dest, minlength=size).cpu().tolist()
    sendbuf = store.get_batch(name, perm[j_send], dtype=dtype)

    # --- recv side: my slots ordered by (source rank, j) to match the
    # concatenation order all_to_all delivers
    src_mine = src_of_slot[p0:p1]
    recv_counts = torch.bincount(src_mine, minlength=size).cpu().tolist()
    order = torch.argsort(src_mine, stable=True)
    recvbuf = torch.empty((p1 - p0, disp), dtype=dtype, device=dev)

    store.comm.all_to_all_single(
        recvbuf,
        sendbuf.view(-1, disp),
        output_split_sizes=recv_counts,
        input_split_sizes=send_counts,
    )

    store._backend.scatter_local(name, order.contiguous(), recvbuf)
    if store.mode == "hip":
        torch.cuda.synchronize(store.device)
    store.comm.barrier()


def _reshuffle_csr(store: DDStore, name: str, seed: int) -> None:
    """CSR reshuffle: samples have variable lengths, so per-rank ELEMENT
    counts change even though sample counts are slot-preserved. Pull-based:
    every rank one-sided-gathers its new samples (striped across xGMI links
    by the random permutation), then the variable is re-registered
    collectively with the new element layout. Transiently holds ~2x the
    shard (old + new) in memory."""
    q = store.query(name)
    meta = store._meta(name)
    dev = store.device
    rank = store.rank
    p0, p1 = int(q["prefix"][rank]), int(q["prefix"][rank + 1])
    ntotal = int(q["nrows_total"])

    g = torch.Generator(device=dev)
    g.manual_seed(int(seed))
    perm = torch.randperm(ntotal, generator=g, device=dev)
    _check_perm_agreement(store, perm)
    mine = perm[p0:p1].contiguous()

    values, _ = store.get_csr(name, mine)
    goff = meta["goff_dev"] if store.mode == "hip" else meta["goff"]
    lens = (goff[mine + 1] - goff[mine]).cpu()
    if store.mode == "hip":
        torch.cuda.synchronize(store.device)
    store.comm.barrier()  # every rank done pulling before shards are freed

    store._backend.free_var(name)
    del store._vars[name]
    store.add_csr(name, values, lens)


def cycle_order_device(perm: torch.Tensor):
    """Concatenated-cycle traversal of a permutation via pointer-doubling
    list ranking -- O(n log n) work but fully parallel, so it runs on the
    GPU in ~seconds where the native sequential walk (dependent random
    loads at DRAM latency) takes ~90 s at 537M rows.

    Produces EXACTLY the same (order, starts) as the serial walk: cycles
    are emitted by ascending minimum element, which is precisely the
    discovery order of a walk that scans start candidates in ascending
    index order; within a cycle the traversal is leader, perm[leader], ...

    Uses int32 working tensors when n < 2^31 (the doubling temps dominate
    transient memory: ~20 GB at 537M rows, fitting beside a 256 GiB shard).
    """
    if isinstance(perm, list):  # ownership transfer: we may free it early
        (perm,) = perm
        owned = True
    else:
        owned = False
    n = perm.numel()
    dev = perm.device
    if n == 0:
        z = torch.zeros(1, dtype=torch.int64)
        return torch.empty(0, dtype=torch.int64, device=dev), z
    idt = torch.int32 if n < 2**31 else torch.int64
    rounds = max(1, int(math.ceil(math.log2(max(n, 2)))))
    chunk = 1 << 26  # bound the i64 index temps beside near-capacity shards

    # leader[e] = min element of e's cycle (min-propagation doubling)
    leader = torch.arange(n, dtype=idt, device=dev)
    jump = perm.to(idt)
    for _ in range(rounds):
        leader = torch.minimum(leader, leader[jump.long()])
        jump = jump[jump.long()]
    del jump

    # distance to the cycle's LAST traversal element (the one whose
    # successor is the leader), via doubling on the leader-broken chain
    succ = perm.to(idt)
    if owned:
        del perm  # free 8n bytes before the ranking peak
    is_last = succ.long() == leader.long()
    d = torch.ones(n, dtype=idt, device=dev)
    d[is_last] = 0
    succ = torch.where(is_last, torch.arange(n, dtype=idt, device=dev), succ)
    del is_last
    for _ in range(rounds):
        sl = succ.long()
        d = d + d[sl]
        succ = succ[sl]
    del succ, sl

    # cycle heads = elements that are their own leader (the cycle minimum);
    # lengths come free from the ranking (d[head] = len-1) -- this avoids a
    # torch.unique whose sort buffers OOM'd beside a 256 GiB shard
    head_mask = leader == torch.arange(n, dtype=idt, device=dev)
    uleaders = torch.nonzero(head_mask, as_tuple=True)[0]  # sorted asc, tiny
    del head_mask
    counts = (d[uleaders] + 1).long()
    starts = torch.zeros(uleaders.numel() + 1, dtype=torch.int64, device=dev)
    torch.cumsum(counts, 0, out=starts[1:])
    # position within cycle = (len-1) - dist_to_last = d[leader] - d, and
    # slot = starts[cycle(e)] + pos(e); both computed in bounded chunks so
    # the unavoidable i64 temps stay ~0.5 GB each (a full-width pass OOM'd
    # beside the 256 GiB shard)
    ul = uleaders.to(idt)
    order = torch.empty(n, dtype=torch.int64, device=dev)
    ar = None
    for lo in range(0, n, chunk):
        hi = min(lo + chunk, n)
        lead_c = leader[lo:hi].long()
        pos_c = (d[lead_c] - d[lo:hi]).long()
        cyc_c = torch.searchsorted(ul, leader[lo:hi]).long()
        if ar is None or ar.numel() != hi - lo:
            ar = torch.arange(lo, hi, dtype=torch.int64, device=dev)
        else:
            ar = torch.arange(lo, hi, dtype=torch.int64, device=dev)
        order[starts[cyc_c] + pos_c] = ar
    return order, starts.cpu()


def reshuffle_epoch_chunked(
    store: DDStore, name: str, seed: int, max_chunk_bytes: int = 1 << 30
) -> None:
    """In-place chunked reshuffle: same slot-preserving result as
    :func:`reshuffle_epoch` (slot j ends up holding the row previously at
    slot perm[j]) but with O(chunk) transient device memory instead of ~2x
    the shard -- required near HBM capacity (BASELINE config 4, ~250 GiB
    shards on a 288 GB GPU).

    Slots are processed in concatenated-CYCLE order (native
    ``cycle_order``): the write to slot order[i] takes the OLD row of slot
    order[i+1] = perm[order[i]], and within a contiguous cycle traversal a
    slot is never read after the position that overwrites it, so chunks of
    the traversal can be applied sequentially with one bounded buffer. The
    one exception is each cycle's CLOSING write, which needs the cycle
    head's original row -- a random permutation has only ~ln(n) cycles, so
    all heads are saved up front. Reads are one-sided batched gathers (the
    store's own xGMI path; random slots stripe all 7 links), a barrier
    separates every chunk's reads from its writes.
    """
    if store._backend.epoch_active():
        raise RuntimeError(
            "ddstore reshuffle: cannot move data inside an open epoch "
            "(concurrent gets would race); call epoch_end() first"
        )
    q = store.query(name)
    if q["is_csr"]:
        raise ValueError(
            "ddstore reshuffle_epoch_chunked: fixed-stride variables only "
            "(CSR shard sizes change; use reshuffle_epoch)"
        )
    from . import _C

    dev = store.device
    rank = store.rank
    ntotal = int(q["nrows_total"])
    disp = int(q["disp"])
    itemsize = int(q["itemsize"])
    dtype = store._meta(name)["dtype"]
    p0, p1 = int(q["prefix"][rank]), int(q["prefix"][rank + 1])
    row_bytes = max(disp * itemsize, 1)

    g = torch.Generator(device=dev)
    g.manual_seed(int(seed))
    perm = torch.randperm(ntotal, generator=g, device=dev)
    _check_perm_agreement(store, perm)
    if store.mode == "hip":
        # parallel list ranking on the GPU (~2 s at 537M rows); the serial
        # host walk is the CPU-mode fallback (~90 s there). Ownership of
        # perm transfers so its 8n bytes free before the ranking peak.
        box = [perm]
        perm_cpu = perm.cpu()  # fallback copy (host RAM)
        del perm
        try:
            order, starts = cycle_order_device(box)
        except torch.OutOfMemoryError:
            # ranking temps did not fit beside the shard: serial host walk
            order, starts = _C.cycle_order(perm_cpu)
        del perm_cpu
    else:
        order, starts = _C.cycle_order(perm.cpu())
        del perm

    odev = order.device  # device-resident on GPU stores (parallel ranking)
    closing_pos = (starts[1:] - 1).to(odev)  # positions that close a cycle
    closing_slot = order[closing_pos]
    heads = order[starts[:-1].to(odev)]  # cycle head slot, aligned
    # save head rows for the closing writes THIS rank owns, before any write
    close_mine = (closing_slot >= p0) & (closing_slot < p1)
    my_heads = heads[close_mine]
    head_rows = (
        store.get_batch(name, my_heads, dtype=dtype)
        if my_heads.numel()
        else None
    )
    head_idx_of_slot = {
        int(s): k for k, s in enumerate(closing_slot[close_mine].tolist())
    }
    if store.mode == "hip":
        torch.cuda.synchronize(dev)
    store.comm.barrier()

    chunk = max(1, int(max_chunk_bytes) // row_bytes)
    owns_all = p0 == 0 and p1 == ntotal  # world_size 1: skip the mask work
    for a in range(0, ntotal, chunk):
        b = min(a + chunk, ntotal)
        W = order[a:b]
        if owns_all:
            w_mine = W
        else:
            mine = (W >= p0) & (W < p1)
            w_mine = W[mine]
        if w_mine.numel():
            pos = (
                torch.arange(a, b, dtype=torch.int64, device=odev)
                if owns_all
                else torch.arange(a, b, dtype=torch.int64, device=odev)[mine]
            )
            is_close = torch.isin(pos, closing_pos)
            src = order[torch.clamp(pos + 1, max=ntotal - 1)]
            # closing writes take the saved head row; give them a harmless
            # in-range source slot for the batched gather, then substitute
            src = torch.where(is_close, w_mine, src)
            buf = store.get_batch(name, src, dtype=dtype)
            if is_close.any():
                rows = is_close.nonzero(as_tuple=True)[0]
                for r in rows.tolist():
                    buf[r] = head_rows[head_idx_of_slot[int(w_mine[r])]]
        if store.mode == "hip":
            torch.cuda.synchronize(dev)
        store.comm.barrier()  # ALL reads of this chunk precede ANY write
        if w_mine.numel():
            lidx = (w_mine - p0).to(dev) if store.mode == "hip" else w_mine - p0
            store._backend.scatter_local(name, lidx.contiguous(), buf)
        if store.mode == "hip":
            torch.cuda.synchronize(dev)
        store.comm.barrier()  # writes land before the next chunk's reads


def expected_perm(ntotal: int, seed: int, device) -> torch.Tensor:
    """The permutation a reshuffle with ``seed`` applies (for tests)."""
    g = torch.Generator(device=device)
    g.manual_seed(int(seed))
    return torch.randperm(ntotal, generator=g, device=device)
