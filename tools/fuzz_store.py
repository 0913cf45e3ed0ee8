#!/usr/bin/env python3
"""Shadow-model fuzzer: drives a DDStore through random operation sequences
and verifies every read against a NumPy mirror. Single rank (the multi-rank
transport has its own tests); catches registry/state-machine/kernel bugs
that hand-written tests miss.

  python tools/fuzz_store.py --ops 500 --seed 0 [--device cpu|cuda]
"""
import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from ddstore_amd import DDStore  # noqa: E402
from ddstore_amd.reshuffle import expected_perm  # noqa: E402

DTYPES = [np.uint8, np.int32, np.int64, np.float32, np.float64]


def run(ops: int, seed: int, device=None, verbose: bool = False) -> int:
    rng = np.random.default_rng(seed)
    store = DDStore(device=device)
    fixed = {}   # name -> np array (current contents)
    csr = {}     # name -> (values np array, lengths np array)
    counter = [0]
    nchecks = 0

    def fresh(prefix):
        counter[0] += 1
        return f"{prefix}{counter[0]}"

    def rand_fixed_arr():
        dt = DTYPES[rng.integers(len(DTYPES))]
        n = int(rng.integers(1, 200))
        d = int(rng.integers(1, 24))
        if dt == np.uint8:
            return rng.integers(0, 255, size=(n, d)).astype(dt)
        if dt in (np.int32, np.int64):
            return rng.integers(-1000, 1000, size=(n, d)).astype(dt)
        return (rng.random((n, d)) * 100).astype(dt)

    for step in range(ops):
        choices = ["add", "add_csr"]
        if fixed:
            choices += ["get", "get_batch", "update", "reshuffle", "free_fixed",
                        "get_batch", "get"]
        if csr:
            choices += ["get_csr", "free_csr"]
        op = choices[rng.integers(len(choices))]

        if op == "add":
            name = fresh("f")
            arr = rand_fixed_arr()
            store.add(name, arr)
            fixed[name] = arr.copy()
        elif op == "add_csr":
            name = fresh("c")
            dt = DTYPES[rng.integers(len(DTYPES))]
            ns = int(rng.integers(1, 60))
            lens = rng.integers(0, 20, size=ns)
            d = int(rng.integers(1, 6))
            vals = (rng.random((max(int(lens.sum()), 1), d)) * 100).astype(dt)[
                : int(lens.sum())
            ]
            store.add_csr(name, vals.reshape(int(lens.sum()), d), lens)
            csr[name] = (vals.copy(), lens.copy(), d)
        elif op == "get":
            name = list(fixed)[rng.integers(len(fixed))]
            arr = fixed[name]
            n = arr.shape[0]
            start = int(rng.integers(0, n))
            count = int(rng.integers(1, n - start + 1))
            out = np.zeros((count, arr.shape[1]), dtype=arr.dtype)
            store.get(name, out, start=start)
            assert np.array_equal(out, arr[start : start + count]), (op, name)
            nchecks += 1
        elif op == "get_batch":
            name = list(fixed)[rng.integers(len(fixed))]
            arr = fixed[name]
            k = int(rng.integers(0, 300))
            idx = rng.integers(0, arr.shape[0], size=k)
            mode = rng.integers(3)
            if mode == 0:       # same dtype (byte move)
                out = store.get_batch(name, idx.astype(np.int64))
                if store.mode == "hip":
                    torch.cuda.synchronize()
                assert np.array_equal(out.cpu().numpy(), arr[idx]), (op, name)
            elif mode == 1:     # fused cast, verified vs torch's own cast
                ot = [torch.float32, torch.float64, torch.int64, torch.bfloat16,
                      torch.float16][rng.integers(5)]
                out = store.get_batch(name, idx.astype(np.int64), dtype=ot)
                if store.mode == "hip":
                    torch.cuda.synchronize()
                ref = torch.from_numpy(arr[idx]).to(ot)
                assert torch.equal(out.cpu(), ref), (op, name, ot)
            else:               # fused affine
                sc = float(rng.uniform(0.1, 3.0))
                sh = float(rng.uniform(-2.0, 2.0))
                out = store.get_batch(name, idx.astype(np.int64),
                                      dtype=torch.float32, affine=(sc, sh))
                if store.mode == "hip":
                    torch.cuda.synchronize()
                ref = (torch.from_numpy(arr[idx]).to(torch.float32) * sc + sh)
                assert torch.allclose(out.cpu(), ref, atol=1e-4, rtol=1e-5), (
                    op, name, sc, sh)
            nchecks += 1
        elif op == "update":
            name = list(fixed)[rng.integers(len(fixed))]
            arr = fixed[name]
            n = arr.shape[0]
            off = int(rng.integers(0, n))
            cnt = int(rng.integers(1, n - off + 1))
            newrows = rand_fixed_arr()[:1].astype(arr.dtype)  # dtype-matched
            newrows = np.broadcast_to(
                newrows[:, :1], (cnt, arr.shape[1])
            ).astype(arr.dtype).copy()
            store.update(name, newrows, offset=off)
            fixed[name][off : off + cnt] = newrows
        elif op == "reshuffle":
            name = list(fixed)[rng.integers(len(fixed))]
            arr = fixed[name]
            sd = int(rng.integers(1 << 30))
            store.reshuffle(name, seed=sd)
            perm = expected_perm(arr.shape[0], sd, store.device).cpu().numpy()
            fixed[name] = arr[perm].copy()
        elif op == "get_csr":
            name = list(csr)[rng.integers(len(csr))]
            vals, lens, d = csr[name]
            goff = np.concatenate([[0], np.cumsum(lens)])
            k = int(rng.integers(0, 100))
            idx = rng.integers(0, len(lens), size=k)
            v, off = store.get_csr(name, idx.astype(np.int64))
            if store.mode == "hip":
                torch.cuda.synchronize()
            off_h = off.cpu().tolist()
            v_h = v.cpu().numpy()
            for j, g in enumerate(idx):
                seg = v_h[off_h[j] : off_h[j + 1]]
                ref = vals[goff[g] : goff[g + 1]]
                assert np.array_equal(seg.ravel(), ref.ravel()), (op, name, j)
            nchecks += 1
        elif op == "free_fixed":
            name = list(fixed)[rng.integers(len(fixed))]
            store._backend.free_var(name)
            del store._vars[name]
            del fixed[name]
        elif op == "free_csr":
            name = list(csr)[rng.integers(len(csr))]
            store._backend.free_var(name)
            del store._vars[name]
            del csr[name]
        if verbose and step % 100 == 0:
            print(f"  step {step}: {len(fixed)} fixed, {len(csr)} csr vars, "
                  f"{nchecks} checks")
    store.free()
    return nchecks


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--ops", type=int, default=500)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--device", default=None)
    args = p.parse_args()
    n = run(args.ops, args.seed, args.device, verbose=True)
    print(f"fuzz OK: {args.ops} ops, {n} verified reads, seed {args.seed}")
