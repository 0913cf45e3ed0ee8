#!/usr/bin/env python3
"""Config-4 capacity demonstration on ONE GPU: build a
near-HBM-capacity shard (default 256 GiB on a 288 GB MI355X), ingest it in
chunks (no full-shard staging tensor), run the timed global-shuffle gather,
then an IN-PLACE chunked reshuffle (O(chunk) transient -- the 2x-transient
all-to-all path cannot run at this size), verify contents, free.

Self-verifying data: row j encodes its id exactly as (col0, col1) =
(j % 2^24, j // 2^24) (f32-exact below 2^24), so after a reshuffle with
seed S, slot j must decode to perm_S[j].

Usage (on the GPU box):
    python tools/capacity_bench.py --gib 256 [--steps 200] [--batch 262144]
Writes one JSON line; redirect into gpurun_out/ and copy to profiles/.
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
from ddstore_amd.reshuffle import expected_perm  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gib", type=float, default=256.0, help="shard size in GiB")
    p.add_argument("--dim", type=int, default=128, help="f32 elems/row (512 B rows)")
    p.add_argument("--batch", type=int, default=262144)
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--fill-chunk-rows", type=int, default=2 * 1024 * 1024)
    p.add_argument("--reshuffle-chunk-gib", type=float, default=1.0)
    p.add_argument("--seed", type=int, default=4242)
    p.add_argument("--skip-reshuffle", action="store_true")
    p.add_argument("--device", default="cuda", help="cpu = small-scale smoke")
    args = p.parse_args()

    use_cuda = args.device.startswith("cuda")
    if use_cuda:
        assert torch.cuda.is_available(), "capacity bench needs the GPU box"
        dev = torch.device("cuda:0")
        torch.cuda.set_device(dev)
    else:
        dev = torch.device("cpu")

    def sync():
        if use_cuda:
            torch.cuda.synchronize()
    row_bytes = args.dim * 4
    rows = int(args.gib * (1 << 30)) // row_bytes
    result = {"tool": "capacity_bench", "rows": rows, "dim": args.dim,
              "shard_GiB": rows * row_bytes / (1 << 30)}

    store = DDStore(device=dev if use_cuda else "cpu")
    t0 = time.perf_counter()
    store.init("big", rows, args.dim, dtype=torch.float32)  # zeroed hipMalloc
    sync()
    result["init_s"] = time.perf_counter() - t0

    # chunked ingest: generate each slice on device, update() into the shard
    # (no host copy, no full-shard staging tensor). Row id j is encoded
    # exactly as (col0, col1) = (j % 2^24, j // 2^24): f32 is exact below
    # 2^24, so slots remain distinguishable at 2^29 rows.
    t0 = time.perf_counter()
    for lo in range(0, rows, args.fill_chunk_rows):
        hi = min(lo + args.fill_chunk_rows, rows)
        ids = torch.arange(lo, hi, dtype=torch.int64, device=dev)
        vals = (
            (ids % (1 << 24))
            .to(torch.float32)
            .unsqueeze(1)
            .repeat(1, args.dim)
        )
        vals[:, 1] = (ids >> 24).to(torch.float32)
        store.update("big", vals, offset=lo)
    del vals, ids
    sync()
    result["fill_s"] = round(time.perf_counter() - t0, 3)
    result["fill_GBps"] = round(rows * row_bytes / result["fill_s"] / 1e9, 1)

    # timed global-shuffle gather at capacity (same shape as bench.py fetch)
    g = torch.Generator(device=dev)
    g.manual_seed(1234)
    buf_dtype = torch.bfloat16 if use_cuda else torch.float32
    nring, nslots = 4, 8
    idx_slots = [
        torch.randint(0, rows, (args.batch,), generator=g, device=dev)
        for _ in range(nslots)
    ]
    bufs = [
        torch.empty(args.batch, args.dim, dtype=buf_dtype, device=dev)
        for _ in range(nring)
    ]
    store.epoch_begin()
    store.get_batch("big", idx_slots[0], out=bufs[0])
    fetch = store.gather_into if use_cuda else (
        lambda n, i, o: store.get_batch(n, i, out=o))
    for k in range(args.warmup):
        fetch("big", idx_slots[k % nslots], bufs[k % nring])
    sync()
    t0 = time.perf_counter()
    for k in range(args.steps):
        fetch("big", idx_slots[k % nslots], bufs[k % nring])
    sync()
    el = time.perf_counter() - t0
    store.epoch_end()
    sps = args.steps * args.batch / el
    result["gather_samples_per_s"] = round(sps, 0)
    result["gather_GBps_read"] = round(sps * row_bytes / 1e9, 1)
    result["gather_us_per_step"] = round(el / args.steps * 1e6, 2)
    # spot-verify gather correctness (f32 fetch of the exact 2-column id)
    probe = idx_slots[0][:4096]
    got = store.get_batch("big", probe, dtype=torch.float32)
    sync()
    expect0 = (probe % (1 << 24)).to(torch.float32)
    expect1 = (probe >> 24).to(torch.float32)
    assert torch.equal(got[:, 0], expect0) and torch.equal(got[:, 1], expect1), \
        "gather returned wrong rows at capacity"
    assert int(store.query("big").get("oob_skipped", 0)) == 0
    del bufs, idx_slots, got

    if not args.skip_reshuffle:
        t0 = time.perf_counter()
        store.reshuffle(
            "big", args.seed,
            max_chunk_bytes=max(int(args.reshuffle_chunk_gib * (1 << 30)), 4096),
        )
        sync()
        result["reshuffle_s"] = round(time.perf_counter() - t0, 2)
        result["reshuffle_GBps_moved"] = round(
            rows * row_bytes / result["reshuffle_s"] / 1e9, 1
        )
        # verify: slot j now holds row perm[j]'s exact 2-column id
        perm = expected_perm(rows, args.seed, dev)
        probe = torch.randint(0, rows, (8192,), device=dev)
        got = store.get_batch("big", probe, dtype=torch.float32)
        sync()
        src = perm[probe]
        nbad = int(
            ((got[:, 0] != (src % (1 << 24)).to(torch.float32))
             | (got[:, 1] != (src >> 24).to(torch.float32))).sum().item()
        )
        result["reshuffle_verify_bad"] = nbad
        assert nbad == 0, f"reshuffle corrupted {nbad}/8192 probed slots"

    free_t0 = time.perf_counter()
    store.free()
    result["free_s"] = round(time.perf_counter() - free_t0, 3)
    if use_cuda:
        mem = torch.cuda.mem_get_info(dev)
        result["hbm_free_after_GiB"] = round(mem[0] / (1 << 30), 1)
        result["hbm_total_GiB"] = round(mem[1] / (1 << 30), 1)
    print(json.dumps(result))


if __name__ == "__main__":
    main()
