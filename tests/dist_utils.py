"""Helpers for multi-process CPU (gloo) distributed tests.

The reference runs distributed tests on a local Ray cluster
(SURVEY.md §4); here we spawn N processes with torch.multiprocessing, each
becoming a rank of a gloo world — exercising the very same mesh/collective
code paths the RCCL ranks run on the GPU box.
"""
from __future__ import annotations

import os
import socket
import traceback
from typing import Callable

import torch.multiprocessing as mp


def _free_port() -> int:
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _to_plain(obj):
    """Tensors -> numpy so queue transport doesn't rely on fd sharing with a
    worker process that may already have exited."""
    import torch
    if torch.is_tensor(obj):
        return obj.detach().cpu().numpy().copy()
    if isinstance(obj, dict):
        return {k: _to_plain(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return type(obj)(_to_plain(v) for v in obj)
    return obj


def _to_torch(obj):
    import numpy as np
    import torch
    if isinstance(obj, np.ndarray):
        return torch.from_numpy(obj)
    if isinstance(obj, dict):
        return {k: _to_torch(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return type(obj)(_to_torch(v) for v in obj)
    return obj


def _worker(local_rank: int, world_size: int, port: int, fn, args, q):
    os.environ["RANK"] = str(local_rank)
    os.environ["LOCAL_RANK"] = str(local_rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        from alpa_amd import init_distributed, shutdown
        init_distributed(backend="gloo")
        result = fn(local_rank, world_size, *args)
        q.put((local_rank, "ok", _to_plain(result)))
        shutdown()
    except Exception:
        q.put((local_rank, "error", traceback.format_exc()))
        raise


def run_distributed(fn: Callable, world_size: int = 2, args=(), timeout=180):
    """Run fn(rank, world_size, *args) in `world_size` processes over gloo.

    Returns the list of per-rank results ordered by rank. Raises on any
    rank failure.
    """
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = []
    port = _free_port()
    for r in range(world_size):
        p = ctx.Process(target=_worker,
                        args=(r, world_size, port, fn, args, q))
        p.start()
        procs.append(p)
    results = {}
    errors = []
    for _ in range(world_size):
        r, status, payload = q.get()
        if status == "error":
            errors.append((r, payload))
        else:
            results[r] = payload
    for p in procs:
        p.join(timeout)
        if p.is_alive():
            p.terminate()
            raise RuntimeError("distributed test worker hung")
    if errors:
        raise RuntimeError("\n".join(f"rank {r}:\n{tb}" for r, tb in errors))
    return [_to_torch(results[r]) for r in range(world_size)]
