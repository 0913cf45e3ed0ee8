#!/usr/bin/env python3
"""Flagship benchmark: global-shuffle sample-fetch throughput of the
MI355X-native store -- the BASELINE.json headline metric ("samples/sec +
effective remote-read GB/s, global shuffle, 1/2/4/8 MI355X"; the reference
publishes no numbers of its own -- BASELINE.md).

Per rank (one process per GPU, RCCL over xGMI for N>1):
  * shard: ``--rows`` rows x ``--dim`` float32 in local HBM (default
    2 Mi x 128 = 1 GiB; 512 B rows -- the reference's per-get granularity,
    test/demo.py:45-50)
  * each timed step fetches ``--batch`` globally-shuffled rows (fraction
    (N-1)/N remote over xGMI) with one CDNA4 gather kernel launch: in-kernel
    owner lookup, peer-pointer reads, pack + fused bf16 cast.
  * ``--mode train`` instead overlaps the fetch (side-stream prefetcher)
    with a full bf16 MLP train step -- forward, backward, optimizer, DDP
    gradient allreduce -- per step (BASELINE config 5).

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Rank 0 prints ONE JSON line: whole-job samples/sec (aggregate over ranks),
plus effective gather + remote-read GB/s in ``extra``.
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--rows", type=int, default=2 * 1024 * 1024, help="rows per rank")
    p.add_argument("--dim", type=int, default=128, help="f32 elems per row (128 -> 512 B rows)")
    p.add_argument("--batch", type=int, default=262144, help="rows fetched per rank per step")
    p.add_argument("--hidden", type=int, default=1024, help="MLP hidden width")
    p.add_argument("--mode", choices=["fetch", "train", "csr"], default="fetch",
                   help="fetch = global-shuffle sample-fetch throughput, the "
                        "BASELINE.json headline metric (default); train = the "
                        "same fetch overlapped with a full bf16 MLP train step "
                        "(forward+backward+optimizer, BASELINE config 5); "
                        "csr = variable-length (HydraGNN-style) sample fetch "
                        "(BASELINE config 3)")
    p.add_argument("--store-dtype", choices=["f32", "bf16", "u8", "fp8"], default="f32",
                   help="shard storage dtype; non-f32 exercises the fused "
                        "expand-on-gather path (fp8 = OCP e4m3fn)")
    p.add_argument("--device", default="cuda")
    p.add_argument("--backend", default=None,
                   help="torch.distributed backend override (default: nccl on "
                        "GPU, gloo on CPU; gloo also works on GPU for "
                        "oversubscribed single-GPU testing)")
    return p.parse_args()


class TrainStep:
    """bf16 MLP forward+backward+SGD consuming a fetched minibatch."""

    def __init__(self, dim: int, hidden: int, device: torch.device, world: int):
        self.model = torch.nn.Sequential(
            torch.nn.Linear(dim, hidden),
            torch.nn.GELU(),
            torch.nn.Linear(hidden, dim),
        ).to(device=device, dtype=torch.bfloat16)
        if world > 1:
            self.model = torch.nn.parallel.DistributedDataParallel(self.model)
        self.opt = torch.optim.SGD(self.model.parameters(), lr=1e-3)

    def __call__(self, batch_bf16: torch.Tensor):
        self.opt.zero_grad(set_to_none=True)
        out = self.model(batch_bf16)
        loss = torch.nn.functional.mse_loss(out, batch_bf16)
        loss.backward()
        self.opt.step()
        return loss


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_cuda = args.device.startswith("cuda") and torch.cuda.is_available()

    if world > 1:
        backend = args.backend or ("nccl" if use_cuda else "gloo")
        dist.init_process_group(backend, rank=rank, world_size=world)
    if use_cuda:
        local_rank = local_rank % max(torch.cuda.device_count(), 1)
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")

    from ddstore_amd import DDStore, PrefetchLoader

    store = DDStore(device=device if use_cuda else "cpu")
    rows, dim, batch = args.rows, args.dim, args.batch
    tdev = device if use_cuda else "cpu"
    sdt = {"f32": torch.float32, "bf16": torch.bfloat16, "u8": torch.uint8,
           "fp8": torch.float8_e4m3fn}[args.store_dtype]
    if args.mode == "csr":
        # variable-length samples: 16..2*dim-16 f32 elements (mean = dim)
        gcpu = torch.Generator().manual_seed(4321 + rank)
        lens = torch.randint(16, 2 * dim - 16, (rows,), generator=gcpu)
        nelems = int(lens.sum())
        shard = torch.randn(nelems, 1, dtype=torch.float32, device=tdev)
        store.add_csr("bench", shard, lens)
    else:
        shard = torch.randn(rows, dim, dtype=torch.float32, device=tdev)
        if sdt == torch.uint8:
            shard = (shard * 64 + 128).clamp(0, 255).to(torch.uint8)
        elif sdt != torch.float32:
            shard = shard.to(sdt)
        store.add("bench", shard)
    del shard

    ntotal = rows * world
    nsteps_total = args.warmup + args.steps
    # sampler: per-epoch global permutation, this rank's disjoint slice
    # (DistributedSampler semantics, reference vae-ddp.py:216). The index
    # pool is capped at 64 Mi rows (512 MB int64) and cycled for very long
    # soaks -- far beyond any cache (64 Mi x 512 B = 32 GiB of rows), so
    # reuse cannot make the fetch cheaper.
    g = torch.Generator().manual_seed(1234)
    need = min(nsteps_total * batch, 64 * 1024 * 1024)
    need = max(need, batch)
    order = []
    got = 0
    while got < need:
        perm = torch.randperm(ntotal, generator=g)
        share = perm[rank::world]  # this rank's slice of the global shuffle
        order.append(share)
        got += share.numel()
    order = torch.cat(order)[:need]

    store.epoch_begin()
    if args.mode == "train":
        # overlap config: side-stream prefetch feeding a full bf16 train step
        trainer = TrainStep(dim, args.hidden, device, world)
        loader = PrefetchLoader(store, "bench", order, batch,
                                out_dtype=torch.bfloat16, depth=3, drop_last=True)
        it = iter(loader)

        def run_steps(n: int):
            for _ in range(n):
                trainer(next(it))

    elif args.mode == "fetch":
        # flagship store metric: back-to-back batched gathers (owner lookup +
        # xGMI peer read + pack + fused bf16 cast per step), nothing else in
        # the timed loop
        order_dev = order.to(device) if use_cuda else order
        nring = 4
        # GPU: fused cast to the bf16 training dtype; CPU compat path: the
        # store dtype (byte move -- the host path has no fused cast)
        buf_dtype = torch.bfloat16 if use_cuda else sdt
        bufs = [
            torch.empty(batch, dim, dtype=buf_dtype, device=device)
            for _ in range(nring)
        ]
        nslots = order_dev.numel() // batch
        step_idx = [
            order_dev[k * batch : (k + 1) * batch].contiguous()
            for k in range(nslots)
        ]
        # one validated get_batch primes dtype/shape checks; the GPU loop
        # then uses the minimal-overhead gather_into path (the CPU compat
        # path casts via get_batch)
        store.get_batch("bench", step_idx[0], out=bufs[0])
        if use_cuda:
            fetch_one = store.gather_into
        else:
            fetch_one = lambda n, i, o: store.get_batch(n, i, out=o)  # noqa: E731
        counter = {"k": 0}

        def run_steps(n: int):
            k = counter["k"]
            for _ in range(n):
                fetch_one("bench", step_idx[k % nslots], bufs[k % nring])
                k += 1
            counter["k"] = k

    if args.mode == "csr":
        # variable-length fetch: capacity ring buffers sized for the worst
        # batch (2*dim elems/sample max), gather_csr per step, no host sync
        order_dev = order.to(device) if use_cuda else order
        nring = 4
        cap = batch * 2 * dim
        bufs = [torch.empty(cap, 1, dtype=torch.float32, device=tdev) for _ in range(nring)]
        nslots = order_dev.numel() // batch
        step_idx = [
            order_dev[k * batch : (k + 1) * batch].contiguous()
            for k in range(nslots)
        ]
        counter = {"k": 0}

        def run_steps(n: int):
            k = counter["k"]
            for _ in range(n):
                store.get_csr("bench", step_idx[k % nslots], out=bufs[k % nring])
                k += 1
            counter["k"] = k

    run_steps(args.warmup)

    if world > 1:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    run_steps(args.steps)
    if use_cuda:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    t1 = time.perf_counter()
    store.epoch_end()

    elapsed = t1 - t0
    if world > 1:
        red_dev = device if (use_cuda and dist.get_backend() == "nccl") else "cpu"
        t = torch.tensor([elapsed], device=red_dev, dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    # integrity: every timed index must have been in range (skipped rows
    # would mean measuring less work than claimed)
    qf = store.query("bench")
    oob = int(qf.get("oob_skipped", 0))
    assert oob == 0, f"bench integrity: {oob} out-of-range indices were skipped"

    n_samples = world * args.steps * batch
    stored_itemsize = {"f32": 4, "bf16": 2, "u8": 1, "fp8": 1}[args.store_dtype]
    row_bytes = dim * (stored_itemsize if args.mode != "csr" else 4)
    sps = n_samples / elapsed
    gather_gbps = n_samples * row_bytes / elapsed / 1e9
    remote_gbps = gather_gbps * (world - 1) / world if world > 0 else 0.0

    if rank == 0:
        result = {
            "metric": {"train": "samples/sec (global-shuffle fetch + bf16 train step)",
                       "fetch": "samples/sec (global-shuffle sample fetch)",
                       "csr": "samples/sec (global-shuffle variable-length fetch)"}[args.mode],
            "value": sps,
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": {"train": f"store-globalshuffle+bf16-mlp{args.hidden}",
                          "fetch": "store-globalshuffle-fetch",
                          "csr": "store-globalshuffle-csr-fetch"}[args.mode],
                "global_batch": world * batch,
                "rows_per_rank": rows,
                "row_bytes": row_bytes,
                "store_dtype": args.store_dtype,
                "shard_GiB": rows * row_bytes / 2**30,
                "parallelism": f"dp{world}",
                "mode": args.mode,
            },
            "extra": {
                "gather_GBps_aggregate": gather_gbps,
                "remote_read_GBps_aggregate": remote_gbps,
                "store_stats_rank0": store.stats().get("bench", {}),
            },
        }
        print(json.dumps(result))

    store.free()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
