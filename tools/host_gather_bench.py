#!/usr/bin/env python3
"""Config-1 host-path tuning evidence: measures the box's
sequential memcpy bandwidth as the roofline, then the HostStore random
gather against it. Random 512-B rows are latency-bound, so the honest
comparison also includes a random torch index_select baseline.

    python tools/host_gather_bench.py [--rows N] [--dim D] [--batch B]
"""
from __future__ import annotations

import argparse
import json
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(
    __import__("os").path.abspath(__file__))))

from ddstore_amd import DDStore  # noqa: E402


def timeit(fn, reps):
    fn()  # warm
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    return (time.perf_counter() - t0) / reps


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rows", type=int, default=4 * 1024 * 1024)
    p.add_argument("--dim", type=int, default=128)
    p.add_argument("--batch", type=int, default=262144)
    p.add_argument("--reps", type=int, default=10)
    args = p.parse_args()
    rows, dim, batch = args.rows, args.dim, args.batch
    row_bytes = dim * 4

    # roofline: sequential big-buffer copy (read+write)
    a = torch.randn(rows, dim)
    b = torch.empty_like(a)
    seq_s = timeit(lambda: b.copy_(a), 3)
    seq_gbps = 2 * rows * row_bytes / seq_s / 1e9  # r+w

    # torch random-gather baseline (same access pattern, same thread pool)
    idx = torch.randint(0, rows, (batch,))
    out_t = torch.empty(batch, dim)
    tsel_s = timeit(lambda: torch.index_select(a, 0, idx, out=out_t), args.reps)
    tsel_gbps = 2 * batch * row_bytes / tsel_s / 1e9

    store = DDStore(device="cpu")
    store.add("h", a)
    del a, b
    out = torch.empty(batch, dim)
    g_s = timeit(lambda: store.get_batch("h", idx, out=out), args.reps)
    g_gbps = 2 * batch * row_bytes / g_s / 1e9
    assert torch.equal(out[:64], store.local_shard("h")[idx[:64]])
    store.free()

    print(json.dumps({
        "tool": "host_gather_bench",
        "threads": torch.get_num_threads(),
        "rows": rows, "dim": dim, "batch": batch,
        "seq_memcpy_GBps_rw": round(seq_gbps, 1),
        "torch_index_select_GBps_rw": round(tsel_gbps, 1),
        "hoststore_gather_GBps_rw": round(g_gbps, 1),
        "gather_vs_memcpy": round(g_gbps / seq_gbps, 3),
        "gather_vs_index_select": round(g_gbps / tsel_gbps, 3),
    }))


if __name__ == "__main__":
    main()
