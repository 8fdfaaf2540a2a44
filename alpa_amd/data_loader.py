"""Distributed data loading: each rank loads exactly its shard.

Capability analog of the reference's ``alpa/data_loader.py``: the driver
version (:15) shards + puts with prefetch; the mesh-worker version (:97)
has every worker load its own shard locally.  In the one-process-per-GPU
runtime every rank IS a worker, so the loader slices the global batch by
this rank's dp coordinate and prefetches to device with a background
stream.
"""
from __future__ import annotations

import collections
import threading
from typing import Any, Callable, Iterable, Iterator, Optional

import torch

from .mesh import DeviceMesh


def shard_batch(batch: Any, dp: int, dp_idx: int) -> Any:
    """Slice every tensor leaf along dim 0 to this dp rank's shard."""
    if torch.is_tensor(batch):
        n = batch.shape[0]
        assert n % dp == 0, (n, dp)
        per = n // dp
        return batch[dp_idx * per:(dp_idx + 1) * per]
    if isinstance(batch, dict):
        return {k: shard_batch(v, dp, dp_idx) for k, v in batch.items()}
    if isinstance(batch, (list, tuple)):
        return type(batch)(shard_batch(v, dp, dp_idx) for v in batch)
    return batch


def _pin(batch: Any) -> Any:
    """Pin host tensors so the H2D copy can actually run async —
    non_blocking from PAGEABLE memory silently degrades to a
    synchronous copy."""
    if torch.is_tensor(batch):
        return batch.pin_memory() if not batch.is_pinned() else batch
    if isinstance(batch, dict):
        return {k: _pin(v) for k, v in batch.items()}
    if isinstance(batch, (list, tuple)):
        return type(batch)(_pin(v) for v in batch)
    return batch


def _to_device(batch: Any, device, non_blocking=True) -> Any:
    if torch.is_tensor(batch):
        return batch.to(device, non_blocking=non_blocking)
    if isinstance(batch, dict):
        return {k: _to_device(v, device, non_blocking)
                for k, v in batch.items()}
    if isinstance(batch, (list, tuple)):
        return type(batch)(_to_device(v, device, non_blocking)
                           for v in batch)
    return batch


class DataLoader:
    """Wraps an iterable of *global* batches; yields this rank's dp shard,
    prefetched to device (reference DataLoader:15 prefetch deque).

    For tp > 1, ranks sharing a dp coordinate receive identical data by
    construction (same slice of the same global batch).
    """

    def __init__(self, batches: Iterable[Any],
                 mesh: Optional[DeviceMesh] = None, dp_axis: int = 0,
                 device: Optional[torch.device] = None,
                 prefetch_size: int = 2):
        self.batches = batches
        self.mesh = mesh
        self.dp = mesh.axis_size(dp_axis) if mesh is not None else 1
        self.dp_idx = max(mesh.axis_index(dp_axis), 0) \
            if (mesh is not None and mesh.is_member) else 0
        self.device = device
        self.prefetch_size = prefetch_size

    def __iter__(self) -> Iterator[Any]:
        queue: collections.deque = collections.deque()
        it = iter(self.batches)
        use_stream = (self.device is not None and
                      torch.cuda.is_available() and
                      torch.device(self.device).type == "cuda")
        copy_stream = torch.cuda.Stream() if use_stream else None

        def load_next():
            try:
                b = next(it)
            except StopIteration:
                return None
            b = shard_batch(b, self.dp, self.dp_idx)
            if self.device is not None:
                if copy_stream is not None:
                    host = _pin(b)
                    with torch.cuda.stream(copy_stream):
                        b = _to_device(host, self.device)
                    ev = torch.cuda.Event()
                    ev.record(copy_stream)
                    # keep the pinned host tensors alive until the copy
                    # is consumed (async H2D reads them after return)
                    return (b, ev, host)
                b = _to_device(b, self.device)
            return (b, None, None)

        for _ in range(self.prefetch_size):
            item = load_next()
            if item is None:
                break
            queue.append(item)
        while queue:
            b, ev, _host = queue.popleft()
            nxt = load_next()
            if nxt is not None:
                queue.append(nxt)
            if ev is not None:
                torch.cuda.current_stream().wait_event(ev)
            yield b


def synthetic_lm_batches(num_batches: int, global_batch: int, seq_len: int,
                         vocab_size: int, seed: int = 0):
    """Deterministic synthetic LM batches (benchmarks; no-network rule)."""
    for i in range(num_batches):
        g = torch.Generator().manual_seed(seed + i)
        yield {
            "ids": torch.randint(0, vocab_size, (global_batch, seq_len),
                                 generator=g),
            "labels": torch.randint(0, vocab_size, (global_batch, seq_len),
                                    generator=g),
        }
