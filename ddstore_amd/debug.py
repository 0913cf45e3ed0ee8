"""Transport-integrity verification (debug aid).

SURVEY §5: the reference has no race detection; its RMA correctness rests on
fence discipline. Here `verify_transport` checks the one-sided path
end-to-end: every rank computes a digest of its OWN shard, digests are
allgathered, then every rank fetches each PEER's full shard through the
normal gather path (xGMI peer reads on GPU, shm on CPU) and compares
digests. Any mismatch means the transport (IPC mapping, directory, gather
kernel) corrupted or raced on data.

Digest: elementwise bit-reinterpret to int64-summable form, summed with
wrap-around -- order-independent, cheap, device-side.
"""
from __future__ import annotations

from typing import Dict, List

import torch

from .store import DDStore


def _digest(t: torch.Tensor) -> int:
    b = t.contiguous().view(torch.uint8).view(-1)
    # pad to 8-byte multiple and fold as int64 with wraparound
    n8 = (b.numel() // 8) * 8
    main = b[:n8].view(torch.int64).sum(dtype=torch.int64)
    tail = b[n8:].to(torch.int64).sum(dtype=torch.int64)
    return int((main + tail).item()) & (2**63 - 1)


def verify_transport(store: DDStore, name: str, chunk_rows: int = 1 << 20) -> Dict:
    """Collective. Returns {'ok': bool, 'mismatches': [(reader, owner)],
    'digests': [...]}; raises nothing on mismatch (caller decides)."""
    q = store.query(name)
    if q["is_csr"]:
        raise NotImplementedError("verify_transport: fixed-stride variables only")
    prefix: List[int] = list(q["prefix"])
    own = _digest(store.local_shard(name))
    digests = store.comm.allgather(own)

    mismatches = []
    for p in range(store.size):
        lo, hi = prefix[p], prefix[p + 1]
        if hi <= lo:
            continue
        acc = 0
        for start in range(lo, hi, chunk_rows):
            stop = min(start + chunk_rows, hi)
            rows = store.get_batch(name, torch.arange(start, stop))
            if store.mode == "hip":
                torch.cuda.synchronize(store.device)
            acc = (acc + _digest(rows)) & (2**63 - 1)
        if acc != digests[p]:
            mismatches.append((store.rank, p))
    all_mism = store.comm.allgather(mismatches)
    flat = [m for ms in all_mism for m in ms]
    return {"ok": not flat, "mismatches": flat, "digests": digests}
