"""Stage construction: choose the number of pipeline stages (and their
submeshes) for a device count + layer cost profile.

Semantics follow the reference's inter-op DP
(``stage_construction.py:235,311`` training_dp): minimize the 1F1B
makespan  ``total_stage_latency + (M - 1) * max_stage_latency``
(Alpa paper eqn. 3) over stage counts and layer clusterings, with
per-stage latency modeled as stage_flops / stage_devices plus the
cross-stage activation transfer on one xGMI link.

This is the cost-model path (reference HloCostModelProfileWorker:414);
profile-guided stage costs can be plugged in via `layer_costs` measured by
the profiling DB (mesh_profiling analog).
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

from ..global_env import global_config
from .layer_clustering import cluster_layers, uniform_layer_costs


def divisors(n: int) -> List[int]:
    return [d for d in range(1, n + 1) if n % d == 0]


def pipeline_makespan(stage_costs: Sequence[float], num_microbatches: int,
                      comm_cost: float = 0.0) -> float:
    """1F1B makespan: fill/drain of all stages + steady state bound by the
    slowest stage (Alpa paper eqn. 3; reference stage_construction.py:289).
    """
    total = sum(stage_costs) + comm_cost * max(0, len(stage_costs) - 1)
    return total + (num_microbatches - 1) * max(stage_costs)


def choose_stages(num_devices: int, num_microbatches: int,
                  layer_costs: Optional[Sequence[float]] = None,
                  num_layers: Optional[int] = None,
                  act_bytes: float = 0.0,
                  max_stages: Optional[int] = None
                  ) -> Tuple[int, List[Tuple[int, int]], float]:
    """Returns (num_stages, layer ranges, estimated makespan).

    Per-stage latency = clustered layer cost * P / num_devices (uniform
    submeshes: more stages => fewer devices per stage => slower stages but
    less intra-stage communication — the model favors P=1 unless memory or
    TP-scaling losses are modeled; callers cap P or provide calibrated
    costs).
    """
    if layer_costs is None:
        assert num_layers is not None
        layer_costs = uniform_layer_costs(num_layers)
    L = len(layer_costs)
    comm = act_bytes * global_config.mesh_beta + global_config.mesh_alpha
    best = None
    for P in divisors(num_devices):
        if P > L:
            break
        if max_stages and P > max_stages:
            break
        ranges = cluster_layers(layer_costs, P)
        per_dev = num_devices // P
        stage_costs = [sum(layer_costs[a:b]) / per_dev for a, b in ranges]
        cost = pipeline_makespan(stage_costs, num_microbatches, comm)
        if best is None or cost < best[2]:
            best = (P, ranges, cost)
    return best


def auto_num_stages(num_devices: int, num_microbatches: int,
                    num_layers: int = 32) -> int:
    return choose_stages(num_devices, num_microbatches,
                         num_layers=num_layers)[0]
