#!/usr/bin/env python3
"""HydraGNN-style training loop over a CSR graph store (BASELINE config 3).

DDStore's purpose in its home project is feeding graph neural network
training from atomistic datasets too large for one node (reference
README.md:200-212). This example shows that loop MI355X-native: each rank
owns a shard of variable-length graphs (node-feature matrices) in HBM, every
epoch draws a DistributedSampler-style global shuffle, minibatches of whole
graphs are packed by the CSR gather kernel (remote graphs over xGMI), pooled
per-graph, and pushed through a bf16 classifier with DDP.

Launch:
  python examples/gnn_csr_train.py --epochs 2
  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      examples/gnn_csr_train.py --epochs 2
"""
import argparse
import os
import sys

import torch
import torch.distributed as dist
import torch.nn as nn

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from ddstore_amd import DDStore, PrefetchLoader  # noqa: E402

FEAT = 16


def segment_mean(values: torch.Tensor, offsets: torch.Tensor) -> torch.Tensor:
    """Mean-pool packed per-node features to per-graph vectors.
    values: (total_nodes, FEAT); offsets: (B+1,) element offsets."""
    B = offsets.numel() - 1
    lens = (offsets[1:] - offsets[:-1]).clamp(min=1)
    gid = torch.repeat_interleave(
        torch.arange(B, device=values.device), offsets[1:] - offsets[:-1]
    )
    pooled = torch.zeros(B, values.shape[1], device=values.device, dtype=values.dtype)
    pooled.index_add_(0, gid, values)
    return pooled / lens.unsqueeze(1).to(values.dtype)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--graphs-per-rank", type=int, default=20000)
    p.add_argument("--batch-size", type=int, default=512)
    p.add_argument("--device", default=None)
    p.add_argument("--backend", default=None,
                   help="dist backend override (default nccl on GPU; use gloo "
                        "to oversubscribe ranks on one GPU)")
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available() if args.device is None else str(
        args.device).startswith("cuda")
    if world > 1:
        backend = args.backend or ("nccl" if use_cuda else "gloo")
        dist.init_process_group(backend, rank=rank, world_size=world)
    if use_cuda:
        local_rank = local_rank % max(torch.cuda.device_count(), 1)
        device = torch.device("cuda", local_rank)
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    store = DDStore(device=device if use_cuda else "cpu")

    # synthetic graphs: class c graphs have node features centered at c
    nloc = args.graphs_per_rank
    g = torch.Generator().manual_seed(rank)
    lens = torch.randint(4, 60, (nloc,), generator=g)
    labels_local = torch.randint(0, 4, (nloc,), generator=g)
    total_nodes = int(lens.sum())
    centers = labels_local.repeat_interleave(lens).to(torch.float32)
    feats = centers.unsqueeze(1) + 0.1 * torch.randn(total_nodes, FEAT, generator=g)
    store.add_csr("graphs", feats, lens)
    store.add("labels", labels_local.unsqueeze(1))

    model = nn.Sequential(
        nn.Linear(FEAT, 128), nn.GELU(), nn.Linear(128, 4)
    ).to(device=device, dtype=torch.bfloat16)
    if world > 1:
        model = nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank] if use_cuda else None)
    opt = torch.optim.Adam(model.parameters(), lr=3e-3)
    lossf = nn.CrossEntropyLoss()

    ntotal = nloc * world
    for epoch in range(args.epochs):
        gs = torch.Generator().manual_seed(100 + epoch)
        mine = torch.randperm(ntotal, generator=gs)[rank::world]
        correct, seen, tot_loss, nb = 0, 0, 0.0, 0
        store.epoch_begin()
        # CSR prefetch: next batch's variable-length gather runs on a side
        # HIP stream while this batch trains
        loader = PrefetchLoader(store, "graphs", mine, args.batch_size,
                                label_name="labels", drop_last=True)
        for (values, offsets), y in loader:
            y = y.view(-1).long()
            x = segment_mean(values[: int(offsets[-1])], offsets).to(torch.bfloat16)
            opt.zero_grad(set_to_none=True)
            logits = model(x)
            loss = lossf(logits.float(), y)
            loss.backward()
            opt.step()
            correct += (logits.argmax(1) == y).sum().item()
            seen += y.numel()
            tot_loss += loss.item()
            nb += 1
        store.epoch_end()
        if rank == 0:
            print(f"epoch {epoch}: loss {tot_loss/max(nb,1):.4f} "
                  f"acc {correct/max(seen,1):.3f} ({nb} batches, mode={store.mode})")
    store.free()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
