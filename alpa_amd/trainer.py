"""High-level training loop helper (reference ``alpa/torch/trainer.py``
``train_torch_module:22``): wraps TrainState + @parallelize + the data
loader into one call for users who don't need a custom loop.

    from alpa_amd.trainer import train_module
    losses = train_module(model_fn, loss_fn, data_iter, method,
                          num_steps=100, lr=1e-4)
"""
from __future__ import annotations

import time
from typing import Any, Callable, Iterable, List, Optional

import torch

from .api import TrainState, parallelize
from .parallel_method import ParallelMethod, ShardParallel


def train_module(model_fn: Callable, loss_fn: Callable,
                 data_iter: Iterable[Any],
                 method: Optional[ParallelMethod] = None,
                 num_steps: Optional[int] = None,
                 lr: float = 1e-4, weight_decay: float = 0.01,
                 log_every: int = 0,
                 state: Optional[TrainState] = None) -> List[float]:
    """Run a training loop and return the per-step losses.

    - ``model_fn(mesh, axis, dtype, device)`` builds the (sharded) model
      — the TrainState.create contract; pass an existing ``state`` to
      continue training instead.
    - ``loss_fn(model, batch)`` returns the scalar microbatch loss.
    - ``data_iter`` yields per-rank batches (each rank its own dp shard;
      tp peers must receive identical batches).
    """
    method = method or ShardParallel()
    if state is None:
        state = TrainState.create(model_fn, method, lr=lr,
                                  weight_decay=weight_decay)
    step = parallelize(loss_fn, method=method)
    losses: List[float] = []
    t0 = time.perf_counter()
    for i, batch in enumerate(data_iter):
        if num_steps is not None and i >= num_steps:
            break
        loss = step(state, batch)
        losses.append(float(loss))
        if log_every and (i + 1) % log_every == 0:
            from .mesh import rank
            if rank() == 0:
                dt = (time.perf_counter() - t0) / (i + 1)
                print(f"step {i + 1}: loss {losses[-1]:.4f} "
                      f"({dt * 1e3:.1f} ms/step avg)")
    return losses


def evaluate_module(state: TrainState, loss_fn: Callable,
                    data_iter: Iterable[Any],
                    num_steps: Optional[int] = None) -> float:
    """Mean eval loss under no_grad with the training placement
    (FollowParallel semantics — reference follow_parallel.py:25)."""
    from .parallel_method import parallelize_inference
    eval_fn = parallelize_inference(loss_fn, state)
    total, n = 0.0, 0
    for i, batch in enumerate(data_iter):
        if num_steps is not None and i >= num_steps:
            break
        total += float(eval_fn(batch))
        n += 1
    return total / max(n, 1)
