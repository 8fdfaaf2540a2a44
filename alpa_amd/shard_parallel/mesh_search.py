"""Logical mesh-shape selection for ShardParallel.

When the user gives no ``logical_mesh_shape``, enumerate the (dp, tp)
factorizations of the world size and pick the cheapest under the alpha-beta
xGMI cost model — the outer loop of the reference's auto-sharding
(``shard_parallel/compile_executable.py`` tries logical shapes; the ILP in
auto_sharding.py scores each).  Until a captured graph is supplied, the
decision uses the closed-form communication model below; with a graph, it
defers to the ILP solver (solver.py).
"""
from __future__ import annotations

from typing import List, Tuple


def factorizations(n: int) -> List[Tuple[int, int]]:
    out = []
    d = 1
    while d <= n:
        if n % d == 0:
            out.append((d, n // d))
        d += 1
    return out


def choose_mesh_shape(method, n: int) -> Tuple[int, int]:
    """ILP-backed (dp, tp) choice when the method carries a model hint;
    pure DP otherwise."""
    if n == 1:
        return (1, 1)
    hint = getattr(method, "model_hint", None)
    if hint and hint.get("family") == "gpt":
        from .auto_sharding import plan_to_logical_shape, solve_gpt_sharding
        budget = hint.get("memory_budget")
        # homogeneous layers: 2 fix the per-layer optimum — but the memory
        # constraint needs the full depth, so keep all layers when a budget
        # is set
        layers = int(hint.get("layers", 2)) if budget else \
            min(int(hint.get("layers", 2)), 2)
        plan = solve_gpt_sharding(
            n, hidden=hint["hidden"], layers=layers,
            vocab=hint["vocab"], tokens=hint["tokens"],
            memory_budget=budget,
            force_data_parallel=method.auto_sharding_option
            .force_data_parallel)
        return plan_to_logical_shape(plan)
    return (n, 1)
