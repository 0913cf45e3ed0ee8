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
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
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

    # fixed-stride, same dtype (uint4 fast path)
    s.add("f32", torch.randn(rows, dim, device=dev))
    idx = torch.randint(0, rows, (B,), device=dev)
    out = torch.empty(B, dim, device=dev)
    el = timeit(lambda: s._backend.gather("f32", idx, out))
    report("gather f32->f32 (512B rows)", el, B * dim * 8)

    # fused cast f32 -> bf16
    outb = torch.empty(B, dim, device=dev, dtype=torch.bfloat16)
    el = timeit(lambda: s._backend.gather("f32", idx, outb))
    report("gather f32->bf16 (fused cast)", el, B * dim * 6)

    # fused cast/expand u8 -> f32
    s.add("u8", torch.randint(0, 255, (rows, dim), device=dev, dtype=torch.uint8))
    outf = torch.empty(B, dim, device=dev, dtype=torch.float32)
    el = timeit(lambda: s._backend.gather("u8", idx, outf))
    report("gather u8->f32 (fused expand)", el, B * dim * 5)

    # fp16 -> bf16
    s.add("f16", torch.randn(rows, dim, device=dev, dtype=torch.float16))
    el = timeit(lambda: s._backend.gather("f16", idx, outb))
    report("gather f16->bf16", el, B * dim * 4)

    # small rows (64 B) -- latency/index-bound regime
    dim_s = 16
    s.add("small", torch.randn(rows, dim_s, device=dev))
    outs = torch.empty(B, dim_s, device=dev)
    el = timeit(lambda: s._backend.gather("small", idx, outs))
    report("gather f32 64B rows", el, B * dim_s * 8)

    # CSR, random lengths averaging ~256 elements x 4B = ~1 KiB samples
    rng = np.random.default_rng(0)
    nsamp = 200000
    lens = rng.integers(32, 480, size=nsamp)
    total = int(lens.sum())
    vals = torch.randn(total, 1, device=dev)
    s.add_csr("csr", vals, lens)
    ncsr = 65536
    cidx = torch.from_numpy(rng.integers(0, nsamp, size=ncsr)).to(dev)
    goff = torch.from_numpy(np.concatenate([[0], np.cumsum(lens)])).to(dev)
    clens = goff[cidx + 1] - goff[cidx]
    out_off = torch.zeros(ncsr + 1, dtype=torch.int64, device=dev)
    torch.cumsum(clens, 0, out=out_off[1:])
    ctotal = int(out_off[-1])
    cout = torch.empty(ctotal, 1, device=dev)
    el = timeit(lambda: s._backend.gather_csr("csr", cidx, out_off, cout, ctotal))
    report(f"gather_csr ~1KiB samples (x{ncsr})", el, ctotal * 8)

    s.free()
    if args.json:
        print(json.dumps(results))
    return results


if __name__ == "__main__":
    sys.exit(0 if main() is not None else 1)
