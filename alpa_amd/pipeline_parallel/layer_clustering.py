"""Auto layer clustering: partition L layers into K contiguous stages.

Algorithm semantics follow the reference's clustering DP
(``layer_construction.py:342`` cluster_jaxpr_by_cost: minimize the max
per-stage compute cost, tie-broken by total imbalance) specialized to a
module-level layer list (our IR is modules, not jaxpr equations).
Device-free; unit-tested on synthetic costs like the reference's
tests/pipeline_parallel/test_layer_construction.py.
"""
from __future__ import annotations

from typing import List, Sequence, Tuple


def cluster_layers(costs: Sequence[float], num_stages: int
                   ) -> List[Tuple[int, int]]:
    """Split layers 0..L-1 into `num_stages` contiguous [start, end) ranges
    minimizing max(range cost), tie-broken by sum of squared costs
    (balance).  O(L^2 * K) DP."""
    L = len(costs)
    K = num_stages
    assert 1 <= K <= L, (K, L)
    prefix = [0.0]
    for c in costs:
        prefix.append(prefix[-1] + c)

    def seg(i, j):  # cost of [i, j)
        return prefix[j] - prefix[i]

    INF = float("inf")
    # f[k][j] = (max_cost, sq_sum) best for first j layers in k stages
    f = [[(INF, INF)] * (L + 1) for _ in range(K + 1)]
    arg = [[-1] * (L + 1) for _ in range(K + 1)]
    f[0][0] = (0.0, 0.0)
    for k in range(1, K + 1):
        for j in range(k, L + 1):
            best = (INF, INF)
            bi = -1
            for i in range(k - 1, j):
                if f[k - 1][i][0] == INF:
                    continue
                s = seg(i, j)
                cand = (max(f[k - 1][i][0], s), f[k - 1][i][1] + s * s)
                if cand < best:
                    best = cand
                    bi = i
            f[k][j] = best
            arg[k][j] = bi
    # backtrack
    ranges = []
    j = L
    for k in range(K, 0, -1):
        i = arg[k][j]
        ranges.append((i, j))
        j = i
    ranges.reverse()
    return ranges


def uniform_layer_costs(num_layers: int) -> List[float]:
    return [1.0] * num_layers
