#!/usr/bin/env python3
"""Multi-rank production-shaped soak: per rank, a fixed-stride shard and a
CSR shard; loop {prefetch epoch -> CSR batch -> reshuffle -> transport
verify} until --seconds elapse. Exercises IPC peers, epoch fences,
all-to-all and the verifier together under sustained load.

  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node N \
      tools/soak_multirank.py --seconds 300 [--backend gloo] [--gib 4]
"""
import argparse
import os
import sys
import time

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from ddstore_amd import DDStore, PrefetchLoader  # noqa: E402
from ddstore_amd.debug import verify_transport  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--seconds", type=float, default=120)
    p.add_argument("--gib", type=float, default=0.5, help="fixed shard GiB/rank")
    p.add_argument("--reshuffle-every", type=int, default=1,
                   help="reshuffle cadence in cycles (0 = never). NB: with a "
                        "gloo backend the all-to-all stages through CPU/TCP "
                        "and is VERY slow for multi-GiB shards; use nccl on "
                        "real multi-GPU nodes or keep shards small")
    p.add_argument("--backend", default=None)
    p.add_argument("--device", default=None)
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_cuda = torch.cuda.is_available() if args.device is None else str(
        args.device).startswith("cuda")
    if world > 1:
        backend = args.backend or ("nccl" if use_cuda else "gloo")
        dist.init_process_group(backend, rank=rank, world_size=world)
    if use_cuda:
        torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))

    store = DDStore(device="cuda" if use_cuda else "cpu")
    dim = 128
    rows = int(args.gib * 2**30 / (dim * 4))
    dev = "cuda" if use_cuda else "cpu"
    store.add("x", torch.randn(rows, dim, device=dev))
    lens = np.random.default_rng(rank).integers(16, 240, size=rows // 16)
    store.add_csr("c", torch.randn(int(lens.sum()), 1, device=dev), lens)

    ntotal = rows * world
    ncsr = (rows // 16) * world
    t_end = time.time() + args.seconds
    cycles = fetched = 0
    while time.time() < t_end:
        g = torch.Generator().manual_seed(cycles)
        mine = torch.randperm(ntotal, generator=g)[rank::world][: 2**20]
        bsz = max(1024, min(131072, mine.numel() // 4))
        with store.epoch():
            for b in PrefetchLoader(store, "x", mine, bsz,
                                    out_dtype=torch.bfloat16, drop_last=True):
                fetched += b.shape[0]
            cidx = torch.randint(0, ncsr, (65536,),
                                 generator=g)
            v, off = store.get_csr("c", cidx)
        if args.reshuffle_every and cycles % args.reshuffle_every == 0:
            store.reshuffle("x", seed=cycles)
        if cycles % 5 == 0:
            r = verify_transport(store, "x")
            assert r["ok"], r
        if use_cuda:
            torch.cuda.synchronize()
        assert store.query("x").get("oob_skipped", 0) == 0
        cycles += 1
        if rank == 0 and cycles % 5 == 0:
            print(f"  cycle {cycles}, {time.time() - (t_end - args.seconds):.0f}s",
                  flush=True)
    if rank == 0:
        print(f"soak OK: world={world} {cycles} cycles, "
              f"{fetched/1e6:.0f}M rows fetched/rank, verifier clean")
    store.free()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
