"""Spawn-based multi-process harness for gloo-backend pipeline tests.

Runs ``fn(rank, world_size, *args)`` in ``world_size`` processes with a
FileStore-backed gloo process group (no TCP rendezvous — robust in
containers where the hostname may not resolve).  Results are returned per
rank; exceptions propagate to the parent.
"""

from __future__ import annotations

import os
import tempfile
import traceback

import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world_size, store_path, fn, args, q):
    try:
        store = dist.FileStore(store_path, world_size)
        dist.init_process_group("gloo", store=store, rank=rank, world_size=world_size)
        out = fn(rank, world_size, *args)
        q.put((rank, "ok", out))
    except Exception:
        q.put((rank, "err", traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def run_dist(world_size: int, fn, *args, timeout: float = 180.0):
    ctx = mp.get_context("spawn")
    with tempfile.TemporaryDirectory() as td:
        store_path = os.path.join(td, "store")
        q = ctx.Queue()
        procs = [
            ctx.Process(target=_worker, args=(r, world_size, store_path, fn, args, q))
            for r in range(world_size)
        ]
        for p in procs:
            p.start()
        results = {}
        try:
            for _ in range(world_size):
                rank, status, payload = q.get(timeout=timeout)
                if status == "err":
                    raise RuntimeError(f"rank {rank} failed:\n{payload}")
                results[rank] = payload
        finally:
            for p in procs:
                p.join(timeout=30)
                if p.is_alive():
                    p.terminate()
    return [results[r] for r in range(world_size)]
