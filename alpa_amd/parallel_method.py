"""Parallel-method strategy objects — the user-visible strategy surface.

Mirrors the reference's ``alpa/parallel_method.py``: `ShardParallel` (:64),
`DataParallel` (:115), `Zero2Parallel` (:130), `Zero3Parallel` (:146),
`PipeshardParallel` (:160), `get_3d_parallel_method` (:247) — re-designed for
a one-process-per-GPU torch.distributed runtime: a method owns the logical
mesh choice (manually given or produced by the auto-sharding ILP), builds
the model over that mesh, and installs the gradient-sync machinery.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Callable, Optional, Tuple

import torch

from .global_env import global_config
from .mesh import (DeviceMesh, full_mesh, get_device_mesh, world_size)


@dataclass
class AutoShardingOption:
    """Options steering the intra-op ILP (reference: auto_sharding.py:48)."""
    force_data_parallel: bool = False
    prefer_reduce_scatter: bool = False
    force_zero_stage_3: bool = False
    allow_all_to_all: bool = True
    force_batch_dim_to_mesh_dim: Optional[int] = None
    memory_budget_per_device: Optional[float] = None


class ParallelMethod:
    """Base class. A method resolves to a concrete (dp, tp) DeviceMesh and
    step-execution policy."""

    num_micro_batches: int = 1

    def resolve_mesh(self) -> DeviceMesh:
        raise NotImplementedError

    # axis roles on the resolved 2-D mesh
    dp_axis: int = 0
    tp_axis: int = 1

    @property
    def zero_stage(self) -> int:
        return 0


@dataclass
class ShardParallel(ParallelMethod):
    """Intra-operator parallelism over a single 2-D logical mesh.

    With ``logical_mesh_shape=None`` the auto-sharding ILP picks the
    (dp, tp) factorization of the world (shard_parallel/auto_sharding.py
    here; reference runs its C++ pass + PuLP ILP, auto_sharding.py:617).
    """
    num_micro_batches: int = 1
    logical_mesh_shape: Optional[Tuple[int, int]] = None
    auto_sharding_option: AutoShardingOption = field(
        default_factory=AutoShardingOption)
    #: model description for the auto-sharding ILP when no manual mesh
    #: shape is given, e.g. {"family": "gpt", "hidden": 2560, "layers": 32,
    #: "vocab": 51200, "tokens": <tokens per microbatch>}
    model_hint: Optional[dict] = None

    def resolve_mesh(self) -> DeviceMesh:
        n = world_size()
        shape = self.logical_mesh_shape
        if shape is None:
            if self.auto_sharding_option.force_data_parallel:
                shape = (n, 1)
            else:
                shape = self._auto_mesh_shape(n)
        assert shape[0] * shape[1] == n, (shape, n)
        return full_mesh(shape)

    def _auto_mesh_shape(self, n: int) -> Tuple[int, int]:
        # Delegated to the ILP-based search (see shard_parallel/).  Import
        # here to keep module load light.
        from .shard_parallel.mesh_search import choose_mesh_shape
        return choose_mesh_shape(self, n)

    @property
    def zero_stage(self) -> int:
        o = self.auto_sharding_option
        if o.force_zero_stage_3:
            return 3
        if o.prefer_reduce_scatter:
            return 2
        return 0


def DataParallel(num_micro_batches: int = 1) -> ShardParallel:
    """Pure DP preset (reference parallel_method.py:115)."""
    return ShardParallel(
        num_micro_batches=num_micro_batches,
        auto_sharding_option=AutoShardingOption(force_data_parallel=True))


def Zero2Parallel(num_micro_batches: int = 1) -> ShardParallel:
    """DP + reduce-scatter grads + sharded optimizer state
    (reference parallel_method.py:130)."""
    return ShardParallel(
        num_micro_batches=num_micro_batches,
        auto_sharding_option=AutoShardingOption(force_data_parallel=True,
                                                prefer_reduce_scatter=True))


def Zero3Parallel(num_micro_batches: int = 1) -> ShardParallel:
    """ZeRO-3: params sharded, all-gathered around use
    (reference parallel_method.py:146)."""
    return ShardParallel(
        num_micro_batches=num_micro_batches,
        auto_sharding_option=AutoShardingOption(force_data_parallel=True,
                                                prefer_reduce_scatter=True,
                                                force_zero_stage_3=True))


@dataclass
class PipeshardParallel(ParallelMethod):
    """Inter-op (pipeline) + intra-op parallelism (reference
    parallel_method.py:160).  Implemented in pipeline_parallel/."""
    num_micro_batches: int = 1
    #: number of pipeline stages; None -> auto stage construction DP
    num_stages: Optional[int] = None
    #: per-stage (dp, tp) logical shape; None -> auto
    stage_mesh_shape: Optional[Tuple[int, int]] = None
    #: HETEROGENEOUS stage meshes: one (dp, tp) per stage, sizes may
    #: differ (the reference's auto-search emits exactly this, e.g.
    #: submeshes (1,2),(1,2),(1,4), suite_auto_gpt.py:63); activations
    #: cross mismatched boundaries through the tile-resharding exchange
    stage_mesh_shapes: Optional[list] = None
    layer_option: str = "auto"  # "auto" | "manual"
    stage_option: str = "uniform"  # "uniform" | "auto" | "manual"
    #: per-device memory budget (bytes) for the auto stage search's
    #: feasibility check (reference max_n_succ_stages)
    memory_budget_per_device: Optional[float] = None
    schedule: str = "1f1b"  # "1f1b" | "gpipe" | "inference"

    def resolve_mesh(self) -> DeviceMesh:
        # The pipeline compiler slices the world into per-stage submeshes;
        # this returns the full world for bookkeeping.
        return full_mesh((1, world_size()))


@dataclass
class CreateStateParallel(ParallelMethod):
    """Build/initialize state directly with the training method's
    placement (reference CreateStateParallel, parallel_method.py:336 +
    create_state_parallel.py:73).  In the SPMD-per-rank runtime,
    TrainState.create already constructs every shard on its target device
    with the resolved mesh — this method makes that alignment explicit by
    borrowing the training method's mesh."""
    train_method: Optional[ParallelMethod] = None

    def resolve_mesh(self) -> DeviceMesh:
        assert self.train_method is not None
        return self.train_method.resolve_mesh()


@dataclass
class FollowParallel(ParallelMethod):
    """Run another function (e.g. an eval/inference step) with the input
    placement of an already-compiled training state (reference
    FollowParallel, parallel_method.py:380 / follow_parallel.py:25)."""
    train_method: Optional[ParallelMethod] = None

    def resolve_mesh(self) -> DeviceMesh:
        assert self.train_method is not None
        return self.train_method.resolve_mesh()


def parallelize_inference(fn, state, num_micro_batches: Optional[int] = None):
    """FollowParallel execution (reference follow_parallel.py:25 /
    parallelize(method=FollowParallel(...))): compile-free reuse of the
    TRAINING state's placement for an eval/inference step.

    - shard-parallel states: fn(model, microbatch) under no_grad with
      the training method's microbatch split; scalar results average
      over microbatches, tensor results concatenate on the batch dim.
    - pipeline states: the engine's INFERENCE schedule (fill-only, no
      backward) drives the stages; last-stage outputs are combined the
      same way and broadcast from the loss rank so every rank returns
      the same value (the reference's output-placement contract).
    """
    import torch as _torch

    nmb = num_micro_batches or getattr(state.method, "num_micro_batches", 1)

    def _combine(outs):
        outs = [o for o in outs if o is not None]
        if not outs:
            return None
        if outs[0].dim() == 0:
            return _torch.stack(outs).mean()
        return _torch.cat(outs, dim=0)

    def run(batch):
        from .api import _split_microbatches
        was_training = state.model.training
        state.model.eval()
        try:
            micro = _split_microbatches(batch, nmb)
            with _torch.no_grad():
                if state.engine is not None:
                    outs = state.engine.inference_step(micro)
                    out = _combine(outs)
                    import torch.distributed as dist
                    from .mesh import is_distributed
                    if is_distributed():
                        if out is None or out.dim() == 0:
                            t = (out if out is not None else
                                 _torch.zeros(())).reshape(1).contiguous()
                            dist.broadcast(t, src=state.engine.loss_src_rank)
                            out = t.reshape(())
                    return out
                return _combine([fn(state.model, mb) for mb in micro])
        finally:
            if was_training:
                state.model.train()

    return run


def get_3d_parallel_method(num_micro_batches: int, data_parallel: int,
                           operator_parallel: int, pipeline_parallel: int
                           ) -> ParallelMethod:
    """Manual DP x TP x PP factorization (reference parallel_method.py:247)."""
    n = world_size()
    if data_parallel == -1:
        data_parallel = n // (operator_parallel * pipeline_parallel)
    assert data_parallel * operator_parallel * pipeline_parallel == n, \
        (data_parallel, operator_parallel, pipeline_parallel, n)
    if pipeline_parallel == 1:
        return ShardParallel(num_micro_batches=num_micro_batches,
                             logical_mesh_shape=(data_parallel,
                                                 operator_parallel))
    return PipeshardParallel(num_micro_batches=num_micro_batches,
                             num_stages=pipeline_parallel,
                             stage_mesh_shape=(data_parallel,
                                               operator_parallel),
                             stage_option="manual")
