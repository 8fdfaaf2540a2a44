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
from ..mesh import (DeviceMesh, device, get_device_mesh,
                    is_distributed, rank, world_size)
from ..parallel.grad_sync import GradSynchronizer
from .layer_clustering import cluster_layers, uniform_layer_costs
from .runtime import PipelineEngine
from .spec import PipelineModelSpec


def resolve_stage_layout(method, n: Optional[int] = None,
                         spec: Optional[PipelineModelSpec] = None):
    """Choose (num_stages, per-stage shapes).  Returns (P, shapes) where
    shapes is a list of one (dp, tp) per stage (heterogeneous allowed).

    With ``stage_option="auto"`` and a spec carrying per-layer flops, the
    profile-guided search (stage_construction.profiled_stage_search)
    picks (P, submesh) from the measured cost DB — the closed
    profiling loop (reference stage_construction training_dp fed by
    HloCostModelProfileWorker)."""
    n = n or world_size()
    if method.stage_mesh_shapes is not None:
        shapes = [tuple(sh) for sh in method.stage_mesh_shapes]
        assert sum(a * b for a, b in shapes) == n, (shapes, n)
        return len(shapes), shapes
    P = method.num_stages
    if P is None:
        if (method.stage_option == "auto" and spec is not None
                and spec.layer_flops is not None
                and method.stage_mesh_shape is None):
            from .stage_construction import (profiled_stage_search,
                                             training_dp_search)
            budget = getattr(method, "memory_budget_per_device", None)
            if budget is None and torch.cuda.is_available():
                # usable fraction of this device's HBM (the reference's
                # XLA memory-fraction knob)
                free, total = torch.cuda.mem_get_info()
                budget = global_config.memory_fraction * total
            got = training_dp_search(
                n, method.num_micro_batches, spec.layer_flops,
                spec.boundary_act_bytes, spec.layer_param_bytes,
                memory_budget=budget,
                stage_cost_curve=spec.stage_cost_curve,
                microbatch_tokens=spec.microbatch_tokens)
            if got is None:  # infeasible under the budget: uniform search
                got = profiled_stage_search(
                    n, method.num_micro_batches, spec.layer_flops,
                    spec.boundary_act_bytes, spec.layer_param_bytes)
            P, shapes, ranges, _cost = got
            # the DP pairs its layer ranges WITH the submeshes — stash
            # them so the compiler does not re-cluster uniformly
            method._auto_layer_ranges = list(ranges)
            return P, shapes
        from .stage_construction import auto_num_stages
        P = auto_num_stages(n, method.num_micro_batches)
    assert n % P == 0, f"world {n} not divisible by {P} stages"
    per = n // P
    shape = method.stage_mesh_shape or (per, 1)
    assert shape[0] * shape[1] == per, (shape, per)
    return P, [shape] * P


def build_pipeline_state(spec: PipelineModelSpec, method, lr: float,
                         betas, weight_decay: float):
    """Constructs (stage_module, engine, optimizer, meshes) for this rank."""
    from ..optim import AdamW

    n = world_size()
    P, stage_shapes = resolve_stage_layout(method, n, spec)
    sizes = [a * b for a, b in stage_shapes]
    starts = [sum(sizes[:s]) for s in range(P)]
    hetero = len(set(stage_shapes)) > 1

    # contiguous rank blocks per stage; all ranks create all stage meshes
    # (group creation is collective)
    stage_meshes = []
    for s in range(P):
        ranks = tuple(range(starts[s], starts[s] + sizes[s]))
        stage_meshes.append(get_device_mesh(ranks, stage_shapes[s]))

    my_rank = rank()
    my_stage = next(s for s in range(P)
                    if starts[s] <= my_rank < starts[s] + sizes[s])
    my_mesh = stage_meshes[my_stage]
    coord = my_mesh.coord
    stage_shape = stage_shapes[my_stage]

    prev_peer = next_peer = None
    if not hetero:
        if my_stage > 0:
            prev_peer = int(stage_meshes[my_stage - 1].grid[coord])
        if my_stage < P - 1:
            next_peer = int(stage_meshes[my_stage + 1].grid[coord])

    # layer clustering (auto DP over costs; reference layer_construction.py:342)
    # — unless the auto stage search already chose ranges paired with
    # its (possibly heterogeneous) submeshes
    ranges = getattr(method, "_auto_layer_ranges", None)
    if ranges is None or len(ranges) != P:
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

    from ..mesh import _get_group
    tied_comms = []

    def _lookup(module, path):
        obj = module
        for attr in path.split("."):
            obj = getattr(obj, attr)
        return obj

    for tg in (spec.tied_groups or []):
        pairs = [((P + s) % P, path) for s, path in tg.items()]
        uniq_stages = sorted({s for s, _ in pairs})
        if len(uniq_stages) < 2:
            # all copies live on one stage: alias the parameters directly
            if my_stage == uniq_stages[0]:
                first = _lookup(stage_module, pairs[0][1])
                for _, path in pairs[1:]:
                    parent_path, _, leaf = path.rpartition(".")
                    parent = _lookup(stage_module, parent_path) \
                        if parent_path else stage_module
                    setattr(parent, leaf, first)
            continue
        coords = [(i, j) for i in range(stage_shape[0])
                  for j in range(stage_shape[1])]
        for (ci, cj) in coords:
            ranks = tuple(int(stage_meshes[si].grid[ci, cj])
                          for si in uniq_stages)
            g = _get_group(ranks)
            if my_rank in ranks and my_stage in uniq_stages:
                for s, path in pairs:
                    if s == my_stage:
                        tied_comms.append(
                            (g, _lookup(stage_module, path)))

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
    if hetero or spec.boundary_parts is not None:
        # heterogeneous boundaries (or non-default activation layouts):
        # activations cross via the tile resharding exchange; the
        # engine slices GLOBAL microbatches by its stage's dp
        engine.hetero = True
        engine.stage_meshes = stage_meshes
        engine.stage_shapes = stage_shapes
        engine.boundary_parts_fn = spec.boundary_parts
    # rank that holds the authoritative loss (first rank of last stage)
    engine.loss_src_rank = int(stage_meshes[P - 1].ranks[0])
    # grad scale divides by the LAST stage's dp (the loss-definition dp);
    # equals every stage's dp in the uniform case
    engine.loss_dp = stage_shapes[-1][0]

    if global_config.collect_trace:
        engine.enable_tracing()
    if global_config.pipeline_check_alive and is_distributed():
        # fail fast (within the dist timeout) if any rank died during
        # stage construction — detection-only, like the reference
        import torch.distributed as dist
        dist.barrier()

    # tied-weight cross-stage allreduce groups (reference N15): one group
    # per (tied set x mesh coordinate); every rank creates every group
    # (collective), members keep (group, local param) pairs
    engine.tied_comms = tied_comms
    return stage_module, engine, opt, gs, stage_meshes
