#!/usr/bin/env python3
"""Functional demo -- parity port of the reference test/demo.py:1-60.

Each rank stores ``--num`` rows x ``--dim`` float64 filled with the constant
rank+1 (self-verifying: a fetched row's mean reveals its owner,
reference demo.py:35-39), then performs ``--nbatch`` epoched random
single-row gets with value checks. Two reference defects are fixed
(SURVEY §2.7): indices span the GLOBAL range (demo.py:47 only ever read
rank 0's shard) and the assert checks the actual owner.

Launch:
  python examples/demo.py                                     # 1 rank
  torchrun --standalone --local-addr 127.0.0.1 \
      --nproc-per-node 4 examples/demo.py [--num N --dim D --nbatch B]
"""
import argparse
import os
import sys

import numpy as np
import torch.distributed as dist

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from ddstore_amd import DDStore  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    # --rows alias: under torchrun, bare --num is ambiguous with
    # torchrun's own --numa-binding (argparse prefix matching)
    p.add_argument("--num", "--rows", dest="num", type=int,
                   default=1024 * 1024)
    p.add_argument("--dim", type=int, default=64)
    p.add_argument("--nbatch", type=int, default=32)
    p.add_argument("--device", default=None)
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world > 1:
        import torch

        backend = "nccl" if (args.device or "").startswith("cuda") or (
            args.device is None and torch.cuda.is_available()
        ) else "gloo"
        dist.init_process_group(backend, rank=rank, world_size=world)

    store = DDStore(device=args.device)
    arr = np.full((args.num, args.dim), rank + 1, dtype=np.float64)
    store.add("demo", arr)
    if rank == 0:
        print(f"[demo] world={world} shard={arr.nbytes / 2**20:.0f} MiB/rank "
              f"mode={store.mode}")

    rng = np.random.default_rng(1000 + rank)
    for b in range(args.nbatch):
        store.epoch_begin()
        idx = int(rng.integers(0, args.num * world))
        out = np.zeros((1, args.dim), dtype=np.float64)
        store.get("demo", out, start=idx)
        expect = idx // args.num + 1
        assert out.mean() == expect, (idx, out.mean(), expect)
        store.epoch_end()
    if rank == 0:
        print(f"[demo] {args.nbatch} epoched random gets verified")
    store.free()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
