"""Pipeshard compile driver: slice the world into stage submeshes, cluster
layers, build this rank's stage + engine.

Capability analog of ``pipeline_parallel/compile_executable.py:48``
(compile_pipeshard_executable): layer clustering -> stage construction ->
per-stage sharding -> runtime wiring, collapsed into module-level stage
construction for the torch runtime.
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from ..global_env import global_config
from ..mesh import DeviceMesh, device, get_device_mesh, rank, world_size
from ..parallel.grad_sync import GradSynchronizer
from .layer_clustering import cluster_layers, uniform_layer_costs
from .runtime import PipelineEngine
from .spec import PipelineModelSpec


def resolve_stage_layout(method, n: Optional[int] = None):
    """Choose (num_stages, per-stage (dp, tp))."""
    n = n or world_size()
    P = method.num_stages
    if P is None:
        from .stage_construction import auto_num_stages
        P = auto_num_stages(n, method.num_micro_batches)
    assert n % P == 0, f"world {n} not divisible by {P} stages"
    per = n // P
    shape = method.stage_mesh_shape or (per, 1)
    assert shape[0] * shape[1] == per, (shape, per)
    return P, shape


def build_pipeline_state(spec: PipelineModelSpec, method, lr: float,
                         betas, weight_decay: float):
    """Constructs (stage_module, engine, optimizer, meshes) for this rank."""
    from ..optim import AdamW

    n = world_size()
    P, stage_shape = resolve_stage_layout(method, n)
    per = n // P

    # contiguous rank blocks per stage; all ranks create all stage meshes
    # (group creation is collective)
    stage_meshes = []
    for s in range(P):
        ranks = tuple(range(s * per, (s + 1) * per))
        stage_meshes.append(get_device_mesh(ranks, stage_shape))

    my_rank = rank()
    my_stage = my_rank // per
    my_mesh = stage_meshes[my_stage]
    coord = my_mesh.coord

    prev_peer = next_peer = None
    if my_stage > 0:
        prev_peer = int(stage_meshes[my_stage - 1].grid[coord])
    if my_stage < P - 1:
        next_peer = int(stage_meshes[my_stage + 1].grid[coord])

    # layer clustering (auto DP over costs; reference layer_construction.py:342)
    costs = spec.layer_costs or uniform_layer_costs(spec.num_layers)
    ranges = cluster_layers(costs, P)
    layer_range = ranges[my_stage]

    dtype = getattr(torch, global_config.compute_dtype) \
        if torch.cuda.is_available() else torch.float32
    stage_module = spec.build_stage(layer_range=layer_range,
                                    is_first=(my_stage == 0),
                                    is_last=(my_stage == P - 1),
                                    mesh=my_mesh, axis=method.tp_axis,
                                    dtype=dtype, device=device())

    gs = GradSynchronizer(list(stage_module.parameters()), my_mesh,
                          axis=method.dp_axis)
    opt = AdamW(stage_module.parameters(), lr=lr, betas=betas,
                weight_decay=weight_decay)

    engine = PipelineEngine(
        stage_module, my_stage, P, my_mesh, prev_peer, next_peer,
        num_microbatches=method.num_micro_batches,
        act_shape=None,  # resolved lazily from the first microbatch
        act_dtype=dtype, schedule=method.schedule, grad_sync=gs)
    engine._act_shape_fn = spec.act_shape
    # rank that holds the authoritative loss (first rank of last stage)
    engine.loss_src_rank = int(stage_meshes[P - 1].ranks[0])
    return stage_module, engine, opt, gs, stage_meshes
