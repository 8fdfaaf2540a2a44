"""Serializable parallel-plan records (reference ``alpa/parallel_plan.py``:
PlacementSpec:13, StagePlan:21, PipelinePlan:33, ParallelPlan:47,
plan_to_method:57).

A solved plan — logical mesh shape, per-node ILP strategy choices, pipeline
stage layout — can be saved to JSON and replayed later without re-running
the solver (the reference's LoadSolutionParallelArgs workflow,
benchmark_parallel_utils.py:39).
"""
from __future__ import annotations

import json
from dataclasses import asdict, dataclass, field
from typing import Dict, List, Optional, Tuple


@dataclass
class PlacementSpec:
    """Where a logical tensor lives: mesh shape + per-dim mesh axes."""
    mesh_shape: Tuple[int, int]
    dim_axes: Tuple[Optional[int], ...]


@dataclass
class StagePlan:
    """Intra-op solution for one (stage's) mesh: the ILP output
    (reference StagePlan includes the solution vector + objective)."""
    logical_mesh_shape: Tuple[int, int]
    strategy_choices: Dict[str, str] = field(default_factory=dict)
    objective: float = 0.0


@dataclass
class PipelinePlan:
    num_stages: int = 1
    schedule: str = "1f1b"
    layer_ranges: List[Tuple[int, int]] = field(default_factory=list)
    stage_mesh_shape: Tuple[int, int] = (1, 1)
    #: heterogeneous per-stage shapes (reference auto-search submeshes);
    #: overrides stage_mesh_shape when set
    stage_mesh_shapes: Optional[List[Tuple[int, int]]] = None


@dataclass
class ParallelPlan:
    world_size: int = 1
    num_micro_batches: int = 1
    pipeline_plan: Optional[PipelinePlan] = None
    stage_plans: List[StagePlan] = field(default_factory=list)

    def save(self, path: str):
        with open(path, "w") as f:
            json.dump(asdict(self), f, indent=2)

    @staticmethod
    def load(path: str) -> "ParallelPlan":
        with open(path) as f:
            d = json.load(f)
        pp = d.get("pipeline_plan")
        return ParallelPlan(
            world_size=d["world_size"],
            num_micro_batches=d["num_micro_batches"],
            pipeline_plan=PipelinePlan(
                num_stages=pp["num_stages"], schedule=pp["schedule"],
                layer_ranges=[tuple(r) for r in pp["layer_ranges"]],
                stage_mesh_shape=tuple(pp["stage_mesh_shape"]),
                stage_mesh_shapes=[tuple(x) for x in
                                   pp["stage_mesh_shapes"]]
                if pp.get("stage_mesh_shapes") else None)
            if pp else None,
            stage_plans=[
                StagePlan(logical_mesh_shape=tuple(s["logical_mesh_shape"]),
                          strategy_choices=s["strategy_choices"],
                          objective=s["objective"])
                for s in d.get("stage_plans", [])
            ])


def plan_to_method(plan: ParallelPlan):
    """Rebuild a ParallelMethod from a saved plan (reference
    plan_to_method, parallel_plan.py:57)."""
    from .parallel_method import PipeshardParallel, ShardParallel
    if plan.pipeline_plan is not None and plan.pipeline_plan.num_stages > 1:
        pp = plan.pipeline_plan
        return PipeshardParallel(
            num_micro_batches=plan.num_micro_batches,
            num_stages=pp.num_stages,
            stage_mesh_shape=tuple(pp.stage_mesh_shape),
            stage_mesh_shapes=pp.stage_mesh_shapes,
            schedule=pp.schedule,
            stage_option="manual")
    shape = plan.stage_plans[0].logical_mesh_shape if plan.stage_plans \
        else (plan.world_size, 1)
    return ShardParallel(num_micro_batches=plan.num_micro_batches,
                         logical_mesh_shape=shape)


def method_to_plan(method, world_size: int,
                   sharding_plan=None) -> ParallelPlan:
    """Record the resolved plan of a method (+ optional ILP output)."""
    from .parallel_method import PipeshardParallel
    if isinstance(method, PipeshardParallel):
        from .pipeline_parallel.compile import resolve_stage_layout
        P, shapes = resolve_stage_layout(method, world_size)
        hetero = len(set(shapes)) > 1
        return ParallelPlan(
            world_size=world_size,
            num_micro_batches=method.num_micro_batches,
            pipeline_plan=PipelinePlan(
                num_stages=P, schedule=method.schedule,
                stage_mesh_shape=tuple(shapes[0]),
                stage_mesh_shapes=[tuple(sh) for sh in shapes]
                if hetero else None))
    sp = StagePlan(
        logical_mesh_shape=method.logical_mesh_shape or (world_size, 1))
    if sharding_plan is not None:
        sp.strategy_choices = dict(sharding_plan.choices)
        sp.objective = sharding_plan.objective
        sp.logical_mesh_shape = sharding_plan.mesh_shape
    return ParallelPlan(world_size=world_size,
                        num_micro_batches=method.num_micro_batches,
                        stage_plans=[sp])
