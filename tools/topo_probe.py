#!/usr/bin/env python3
"""xGMI/peer topology probe for multi-GPU boxes (round-2 diagnostics).

Launched one rank per GPU:
  torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
      tools/topo_probe.py

Rank 0 prints, for every (reader, owner) pair, the effective bandwidth of a
pure remote gather (all indices in `owner`'s shard) through the store's IPC
peer-pointer path -- i.e., the per-xGMI-link rate the SCALE bench ultimately
rides on -- plus the all-peers random-gather aggregate per reader.
"""
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from ddstore_amd import DDStore  # noqa: E402


def main():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world > 1:
        dist.init_process_group("nccl", rank=rank, world_size=world)
    torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))

    store = DDStore()
    rows, dim = 1 << 20, 128  # 512 MiB shard, 512 B rows
    store.add("probe", torch.randn(rows, dim, device="cuda"))

    B = 1 << 18
    out = torch.empty(B, dim, dtype=torch.bfloat16, device="cuda")

    def bw_for(idx):
        idx = idx.cuda()
        for _ in range(3):
            store.gather_into("probe", idx, out)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            store.gather_into("probe", idx, out)
        torch.cuda.synchronize()
        el = (time.perf_counter() - t0) / 10
        return B * dim * 4 / el / 1e9  # read-side GB/s

    # pairwise: all reads target one owner's shard
    table = []
    for owner in range(world):
        idx = torch.randint(0, rows, (B,), dtype=torch.int64) + owner * rows
        if world > 1:
            dist.barrier()
        table.append(bw_for(idx))
    # aggregate: random across all owners
    if world > 1:
        dist.barrier()
    agg = bw_for(torch.randint(0, rows * world, (B,), dtype=torch.int64))

    if world > 1:
        rows_all = [None] * world
        dist.all_gather_object(rows_all, (table, agg))
    else:
        rows_all = [(table, agg)]
    if rank == 0:
        print(f"read-side GB/s per (reader <- owner), B={B} 512 B rows:")
        hdr = "         " + "".join(f" own{o:>2d} " for o in range(world))
        print(hdr + "   | all-random")
        for r, (t, a) in enumerate(rows_all):
            line = f"reader {r:2d}" + "".join(f" {v:6.0f}" for v in t)
            print(line + f"   | {a:6.0f}")
        print("(diagonal = local HBM; off-diagonal = one xGMI link, "
              "expect ~150 GB/s read-side per link; all-random stripes all "
              "links, expect ~(N-1)/N * link-aggregate + local share)")
    store.free()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
