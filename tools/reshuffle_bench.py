#!/usr/bin/env python3
"""Epoch all-to-all reshuffle benchmark (BASELINE config 4).

Measures the data-movement rate of ``reshuffle_epoch`` -- gather into send
order, all-to-all(v) over RCCL/xGMI, scatter into slot order -- for a store
sharded across N ranks. On one node of 8 MI355X the all-to-all stripes
pairwise over the 7 xGMI links per GPU.

Launch:
  python tools/reshuffle_bench.py --gib 4                    # 1 rank
  torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
      tools/reshuffle_bench.py --gib 32     # ~256 GiB store across the node
"""
import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from ddstore_amd import DDStore  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gib", type=float, default=4.0, help="shard GiB per rank")
    p.add_argument("--dim", type=int, default=256, help="f32 elems per row (1 KiB rows)")
    p.add_argument("--iters", type=int, default=3)
    p.add_argument("--device", default=None)
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_cuda = torch.cuda.is_available() if args.device is None else str(
        args.device).startswith("cuda")
    if world > 1:
        dist.init_process_group("nccl" if use_cuda else "gloo",
                                rank=rank, world_size=world)
    if use_cuda:
        torch.cuda.set_device(local_rank)

    dev = torch.device("cuda", local_rank) if use_cuda else "cpu"
    store = DDStore(device=dev)
    row_bytes = args.dim * 4
    rows = int(args.gib * 2**30 / row_bytes)
    shard = torch.randn(rows, args.dim, device=dev if use_cuda else "cpu")
    store.add("r", shard)
    del shard

    times = []
    for i in range(args.iters):
        if world > 1:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        store.reshuffle("r", seed=1000 + i)
        if use_cuda:
            torch.cuda.synchronize()
        if world > 1:
            dist.barrier()
        times.append(time.perf_counter() - t0)

    el = min(times)
    store_bytes = world * rows * row_bytes
    if rank == 0:
        print(json.dumps({
            "metric": "reshuffle GB/s (whole-store epoch all-to-all)",
            "n_gpus": world,
            "store_GiB": store_bytes / 2**30,
            "epoch_time_s": el,
            "GBps": store_bytes / el / 1e9,
            "remote_fraction": (world - 1) / world,
            "iters": args.iters,
        }))
    store.free()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
