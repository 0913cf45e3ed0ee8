#!/usr/bin/env python3
"""Kernel microbenchmarks for the store hot path (run on an MI355X box).

Reports effective r+w bandwidth of the gather/pack kernels against the
~6.3 TB/s achievable HBM ceiling (MI355X_MICROARCH.md). Used to generate the
committed evidence in profiles/.
"""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))


def timeit(fn, n=100, warmup=10):
    """fn(i) is called with a rotating iteration index so callers can rotate
    index/output buffers -- reusing ONE index buffer lets the gathered rows
    go L3-resident (256 MiB Infinity Cache) and inflates bandwidth ~30%."""
    for i in range(warmup):
        fn(i)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(n):
        fn(i)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rows", type=int, default=2 * 1024 * 1024)
    p.add_argument("--dim", type=int, default=128)
    p.add_argument("--batch", type=int, default=131072)
    p.add_argument("--json", action="store_true")
    args = p.parse_args()

    from ddstore_amd import DDStore

    dev = torch.device("cuda:0")
    s = DDStore(device=dev)
    rows, dim, B = args.rows, args.dim, args.batch
    results = {}

    def report(name, el, bytes_moved):
        gbps = bytes_moved / el / 1e9
        results[name] = {"us": el * 1e6, "GBps": gbps}
        if not args.json:
            print(f"{name:38s} {el*1e6:8.1f} us   {gbps:7.0f} GB/s (r+w)")

    NIDX = 8  # rotate fresh random index sets so rows are not L3-resident
    idxs = [torch.randint(0, rows, (B,), device=dev) for _ in range(NIDX)]

    # fixed-stride, same dtype (uint4 fast path)
    s.add("f32", torch.randn(rows, dim, device=dev))
    out = torch.empty(B, dim, device=dev)
    el = timeit(lambda i: s._backend.gather("f32", idxs[i % NIDX], out))
    report("gather f32->f32 (512B rows)", el, B * dim * 8)

    # the obvious PyTorch alternative for the same operation
    el = timeit(lambda i: torch.index_select(s.local_shard("f32"), 0, idxs[i % NIDX]))
    report("torch.index_select f32 (baseline)", el, B * dim * 8)

    # sorted indices (same row SET, ascending order): DRAM-locality probe --
    # legal for training since a shuffled batch is order-invariant
    sidxs = [torch.sort(i)[0].contiguous() for i in idxs]
    el = timeit(lambda i: s._backend.gather("f32", sidxs[i % NIDX], out))
    report("gather f32 512B rows, sorted idx", el, B * dim * 8)

    # fused cast f32 -> bf16
    outb = torch.empty(B, dim, device=dev, dtype=torch.bfloat16)
    el = timeit(lambda i: s._backend.gather("f32", idxs[i % NIDX], outb))
    report("gather f32->bf16 (fused cast)", el, B * dim * 6)

    # fused cast/expand u8 -> f32
    s.add("u8", torch.randint(0, 255, (rows, dim), device=dev, dtype=torch.uint8))
    outf = torch.empty(B, dim, device=dev, dtype=torch.float32)
    el = timeit(lambda i: s._backend.gather("u8", idxs[i % NIDX], outf))
    report("gather u8->f32 (fused expand)", el, B * dim * 5)

    # fp16 -> bf16
    s.add("f16", torch.randn(rows, dim, device=dev, dtype=torch.float16))
    el = timeit(lambda i: s._backend.gather("f16", idxs[i % NIDX], outb))
    report("gather f16->bf16", el, B * dim * 4)

    # small rows (64 B) -- latency/index-bound regime
    dim_s = 16
    s.add("small", torch.randn(rows, dim_s, device=dev))
    outs = torch.empty(B, dim_s, device=dev)
    el = timeit(lambda i: s._backend.gather("small", idxs[i % NIDX], outs))
    report("gather f32 64B rows", el, B * dim_s * 8)

    # CSR, random lengths averaging ~256 elements x 4B = ~1 KiB samples
    rng = np.random.default_rng(0)
    nsamp = 200000
    lens = rng.integers(32, 480, size=nsamp)
    total = int(lens.sum())
    vals = torch.randn(total, 1, device=dev)
    s.add_csr("csr", vals, lens)
    ncsr = 65536
    goff = torch.from_numpy(np.concatenate([[0], np.cumsum(lens)])).to(dev)
    plans = []
    capacity = ncsr * 480
    cout = torch.empty(capacity, 1, device=dev)
    for _ in range(NIDX):
        cidx = torch.from_numpy(rng.integers(0, nsamp, size=ncsr)).to(dev)
        clens = goff[cidx + 1] - goff[cidx]
        out_off = torch.zeros(ncsr + 1, dtype=torch.int64, device=dev)
        torch.cumsum(clens, 0, out=out_off[1:])
        plans.append((cidx, out_off, int(out_off[-1])))
    el = timeit(lambda i: s._backend.gather_csr(
        "csr", plans[i % NIDX][0], plans[i % NIDX][1], cout, plans[i % NIDX][2]))
    avg_total = sum(p[2] for p in plans) / NIDX
    report(f"gather_csr ~1KiB samples (x{ncsr})", el, avg_total * 8)

    # dense get_range (the reference's core read primitive): big contiguous
    # block, device-to-device through the peer pointer path
    rout = torch.empty(1 << 16, dim, device=dev)
    el = timeit(lambda i: s._backend.get_range("f32", (i % 8) << 16, 1 << 16, rout))
    report("get_range 32 MiB dense block (D2D)", el, (1 << 16) * dim * 8)

    # reference demo workload: single 512 B row per get, synchronous to host
    # (test/demo.py:45-50 pattern) -- per-op latency, not bandwidth
    import numpy as np2
    one = np2.zeros((1, dim), dtype=np2.float32)
    t0 = time.perf_counter()
    NGET = 200
    for i in range(NGET):
        s.get("f32", one, start=(i * 7919) % rows)
    el = (time.perf_counter() - t0) / NGET
    if not args.json:
        print(f"{'compat get() 1 row, sync D2H':38s} {el*1e6:8.1f} us/get")
    results["compat_get_1row_us"] = {"us": el * 1e6, "GBps": dim * 4 / el / 1e9}

    s.free()
    if args.json:
        print(json.dumps(results))
    return results


if __name__ == "__main__":
    sys.exit(0 if main() is not None else 1)
