"""Helpers to run a test function in N spawned torch.distributed processes.

The reference tests its distributed behavior by launching the same script
under ``mpirun -n 4``; here the same effect comes from spawning
ranks with a gloo (CPU) or gloo-control-plane (GPU) process group on
127.0.0.1.
"""
from __future__ import annotations

import os
import socket
import traceback

import torch.multiprocessing as mp


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _entry(fn, rank, world, port, backend, args, q):
    try:
        import torch.distributed as dist

        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group(backend, rank=rank, world_size=world)
        try:
            fn(rank, world, *args)
        finally:
            if dist.is_initialized():
                dist.destroy_process_group()
        q.put((rank, None))
    except Exception:
        q.put((rank, traceback.format_exc()))


def run_dist(fn, world_size: int, *args, backend: str = "gloo", timeout: float = 180.0):
    """Run ``fn(rank, world_size, *args)`` in ``world_size`` spawned processes
    with an initialized torch.distributed group. Raises on any rank failure."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(target=_entry, args=(fn, r, world_size, port, backend, args, q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    errors = []
    for _ in range(world_size):
        rank, err = q.get(timeout=timeout)
        if err is not None:
            errors.append(f"rank {rank}:\n{err}")
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
            errors.append("a rank did not exit")
    if errors:
        raise AssertionError("\n".join(errors))
