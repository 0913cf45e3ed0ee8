"""Side-stream minibatch prefetcher.

The reference has no prefetch: its DataLoader path issues one blocking
MPI_Get per sample (reference call stack SURVEY §3.2/§3.3). Here the next
minibatch is gathered by the CDNA4 gather kernel on a dedicated side HIP
stream, double-buffered, so the training step on the main stream never waits
for sample fetch (BASELINE config 5). Synchronization is via HIP events:

    side stream:  [wait buf free] -> gather batch k -> record ready_k
    main stream:  wait ready_k -> train step on batch k -> record free_k

Allocator note: the data/label ring buffers are allocated once and reused,
so the caching allocator never recycles them mid-flight. CSR offset tensors
ARE allocated per batch on a side stream; their memory can only be
recycled by a later allocation on the SAME side stream, every launch first
waits ``side.wait_event(free_k)`` for some k' >= the batch whose tensors
died, and the consumer records its free events in order on the single main
stream after all its reads up to that batch -- so with one side stream or
the rotating pair, every reuse is event-ordered after every read.
"""
from __future__ import annotations

import os
from typing import Iterator, Optional, Sequence, Union

import numpy as np
import torch

from .store import DDStore


class PrefetchLoader:
    """Iterate minibatches of ``store[name]`` (and optionally a second
    aligned variable, e.g. labels) for a given epoch index order.

    Parameters
    ----------
    store : DDStore
    name : str            main (data) variable
    indices : 1-D int sequence/tensor of global row ids for this epoch
              (e.g. the ids a DistributedSampler assigned to this rank)
    batch_size : int      rows per yielded batch
    out_dtype : optional  fused dtype cast in the gather kernel
    label_name : optional second variable gathered with the same indices
    depth : int           ring depth (2 = classic double buffering)
    drop_last : bool
    """

    def __init__(
        self,
        store: DDStore,
        name: str,
        indices: Union[Sequence[int], torch.Tensor, np.ndarray],
        batch_size: int,
        out_dtype: Optional[torch.dtype] = None,
        label_name: Optional[str] = None,
        label_dtype: Optional[torch.dtype] = None,
        depth: Optional[int] = None,
        drop_last: bool = False,
        affine: Optional[tuple] = None,
        collect_events: bool = False,
    ):
        # collect_events=True records (start, end) CUDA events around every
        # side-stream fetch into self.fetch_events -- overlap instrumentation
        # (the end-to-end wall-clock delta cannot resolve a ~100-us fetch
        # under a multi-ms train step against timing noise)
        self.collect_events = collect_events
        self.fetch_events: list = []
        self.affine = affine
        if depth is None:
            depth = int(os.environ.get("DDSTORE_PREFETCH_DEPTH", "2"))
        self.store = store
        self.name = name
        self.label_name = label_name
        self.out_dtype = out_dtype
        self.label_dtype = label_dtype
        self.batch_size = int(batch_size)
        self.depth = max(2, int(depth))
        self.drop_last = drop_last
        idx = torch.as_tensor(indices, dtype=torch.int64).flatten()
        self.indices = idx
        self.is_csr = store._meta(name)["is_csr"]
        if self.is_csr:
            if affine is not None:
                raise ValueError("affine is not supported for CSR variables")
            if label_name is not None and store._meta(label_name)["is_csr"]:
                raise ValueError("label variable must be fixed-stride")
            # capacity ring buffers sized for the worst batch
            goff = store._meta(name)["goff"]
            self._max_len = int((goff[1:] - goff[:-1]).max().item()) if goff.numel() > 1 else 0

    def __len__(self) -> int:
        n = self.indices.numel()
        return n // self.batch_size if self.drop_last else (n + self.batch_size - 1) // self.batch_size

    def _batches(self, idx: torch.Tensor):
        n = idx.numel()
        stop = (n // self.batch_size) * self.batch_size if self.drop_last else n
        return [idx[i : min(i + self.batch_size, stop)] for i in range(0, stop, self.batch_size)]

    def _fetch(self, b: torch.Tensor, slot: Optional[dict] = None):
        """One batch fetch; CSR variables yield (values, offsets) tuples."""
        store = self.store
        if self.is_csr:
            cap = b.numel() * self._max_len
            buf = None if slot is None else slot.get("data")
            if buf is None or buf.shape[0] < cap:
                buf = torch.empty(
                    (cap, store._meta(self.name)["disp"]),
                    dtype=store._meta(self.name)["dtype"], device=store.device,
                )
            data = store.get_csr(self.name, b, out=buf)  # (values, offsets)
        else:
            buf = None if slot is None else slot.get("data")
            if buf is not None and buf.shape[0] != b.numel():
                buf = None  # ragged final batch: allocate a matching buffer
            if self.affine is not None and buf is None:
                buf = torch.empty((b.numel(), store._meta(self.name)["disp"]),
                                  dtype=self.out_dtype or torch.float32,
                                  device=store.device)
            data = store.get_batch(self.name, b, out=buf, dtype=self.out_dtype,
                                   affine=self.affine)
        if slot is not None:
            slot["data"] = data[0] if self.is_csr else data
        if self.label_name is None:
            return data, None
        lbuf = None if slot is None else slot.get("label")
        if lbuf is not None and lbuf.shape[0] != b.numel():
            lbuf = None
        label = store.get_batch(self.label_name, b, out=lbuf, dtype=self.label_dtype)
        if slot is not None:
            slot["label"] = label
        return data, label

    def __iter__(self) -> Iterator:
        if self.store.mode != "hip":
            for b in self._batches(self.indices):
                data, label = self._fetch(b)
                yield data if label is None else (data, label)
            return
        yield from self._iter_hip()

    def _iter_hip(self) -> Iterator:
        store = self.store
        device = store.device
        # two rotating side streams: batch k+1's plan/small kernels pipeline
        # under batch k's payload gather (measured +12% on the CSR fetch
        # bench; per-slot event choreography makes the rotation safe)
        sides = [torch.cuda.Stream(device), torch.cuda.Stream(device)]
        idx_dev = self.indices.to(device, non_blocking=False)
        batches = self._batches(idx_dev)
        nb = len(batches)
        slots = [
            {
                "data": None,
                "label": None,
                "ready": torch.cuda.Event(),
                "free": torch.cuda.Event(),
                "free_recorded": False,
            }
            for _ in range(self.depth)
        ]

        def launch(j: int):
            slot = slots[j % self.depth]
            side = sides[j % 2]
            with torch.cuda.stream(side):
                if slot["free_recorded"]:
                    side.wait_event(slot["free"])
                if self.collect_events:
                    ev_s = torch.cuda.Event(enable_timing=True)
                    ev_s.record(side)
                data, label = self._fetch(batches[j], slot)
                slot["yield"] = data if label is None else (data, label)
                slot["ready"].record(side)
                if self.collect_events:
                    ev_e = torch.cuda.Event(enable_timing=True)
                    ev_e.record(side)
                    self.fetch_events.append((ev_s, ev_e))

        for j in range(min(self.depth, nb)):
            launch(j)
        cur = torch.cuda.current_stream(device)
        for i in range(nb):
            slot = slots[i % self.depth]
            cur = torch.cuda.current_stream(device)
            cur.wait_event(slot["ready"])
            yield slot["yield"]
            # the consumer has enqueued its use of the buffers on the current
            # stream by the time it asks for the next batch
            cur = torch.cuda.current_stream(device)
            slot["free"].record(cur)
            slot["free_recorded"] = True
            nxt = i + self.depth
            if nxt < nb:
                launch(nxt)
