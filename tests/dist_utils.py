"""Helpers to run multi-process gloo tests (world_size=2) on CPU."""

from __future__ import annotations

import os
import socket

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _entry(rank: int, world: int, port: int, fn, args, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        out = fn(rank, world, *args)
        q.put((rank, "ok", out))
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put((rank, "err", f"{e}\n{traceback.format_exc()}"))
        raise
    finally:
        dist.destroy_process_group()
        for k in ("RANK", "LOCAL_RANK", "WORLD_SIZE", "MASTER_ADDR", "MASTER_PORT"):
            os.environ.pop(k, None)


def run_distributed(fn, world: int = 2, args: tuple = (), timeout: int = 180):
    """Spawn `world` gloo ranks running fn(rank, world, *args); returns
    {rank: result}. Raises on any rank failure."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = free_port()
    procs = [
        ctx.Process(target=_entry, args=(r, world, port, fn, args, q))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, status, out = q.get()
        if status == "err":
            for p in procs:
                p.terminate()
            raise RuntimeError(f"rank {rank} failed:\n{out}")
        results[rank] = out
    for p in procs:
        p.join(timeout)
    return results
