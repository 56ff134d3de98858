"""Helpers to run a test function across N gloo processes (CPU)."""

from __future__ import annotations

import os
import traceback

import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world_size, port, fn, args, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world_size)
        result = fn(rank, world_size, *args)
        # serialize tensors to plain bytes — shared-memory FD transport
        # races with worker exit (EOFError in the parent)
        import io
        import torch
        buf = io.BytesIO()
        torch.save(result, buf)
        q.put((rank, "ok", buf.getvalue()))
    except Exception:
        q.put((rank, "err", traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _free_port() -> int:
    # bind-to-0 gives an OS-assigned free port (safe under parallel test
    # runners; the tiny close→gloo-bind race is acceptable for CI)
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def run_distributed(fn, world_size: int, *args):
    """Run fn(rank, world_size, *args) in world_size gloo processes.
    Returns list of per-rank results ordered by rank. Raises on any failure."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [
        ctx.Process(target=_worker, args=(r, world_size, port, fn, args, q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world_size):
        rank, status, payload = q.get()
        if status == "err":
            for p in procs:
                p.terminate()
            raise RuntimeError(f"rank {rank} failed:\n{payload}")
        import io
        import torch
        results[rank] = torch.load(io.BytesIO(payload), weights_only=False)
    for p in procs:
        p.join(timeout=60)
    return [results[r] for r in range(world_size)]
