"""The intra-op auto-sharding ILP, solved with HiGHS (scipy.optimize.milp).

Formulation mirrors the reference's ``_call_solver_serialized_args``
(auto_sharding.py:617-872):
  binary s[i][r]  — node i uses strategy r
  binary e[i,j][r,c] — edge (i,j) uses the (r,c) resharding entry
  min  Σ_i s[i]·(compute + comm) + Σ_(i,j) e[i,j]·resharding
  s.t. Σ_r s[i][r] = 1                      (one-hot, :777)
       Σ_rc e[i,j][r,c] = 1
       Σ_c e[r,c] <= s_i[r],  Σ_r e[r,c] <= s_j[c]   (consistency, :794)
       Σ_i  s[i]·mem[i]  <= memory_budget            (memory, :781)
  follow-lists (s_follow, :716) eliminate variables for nodes that
  inherit their producer's strategy.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import numpy as np
from scipy.optimize import Bounds, LinearConstraint, milp
from scipy.sparse import lil_matrix

from .ir import Graph, MeshModel, Node

INFINITY_COST = 1e13  # reference auto_sharding.py:45


@dataclass
class SolveResult:
    objective: float
    choices: Dict[int, int]        # node id -> strategy index
    feasible: bool


def _root(graph: Graph, i: int) -> int:
    """Resolve follow chains to the owning node."""
    seen = set()
    while graph.nodes[i].follow is not None:
        assert i not in seen
        seen.add(i)
        i = graph.nodes[i].follow
    return i


def solve(graph: Graph, mesh: MeshModel,
          memory_budget: Optional[float] = None,
          time_limit: float = 60.0,
          edge_cost_factor: float = 2.0) -> SolveResult:
    """edge_cost_factor=2 for training graphs: every forward resharding has
    a mirrored backward transfer of the gradient (all-gather <-> reduce-
    scatter pair); 1.0 for inference."""
    nodes = graph.nodes
    # owners = nodes with their own strategy variable
    owners = [i for i, n in enumerate(nodes) if n.follow is None]
    owner_of = {i: _root(graph, i) for i in range(len(nodes))}
    nstrat = {i: len(nodes[i].strategies) for i in owners}
    for i in owners:
        assert nstrat[i] > 0, f"node {nodes[i].name} has no strategies"

    # variable layout: all s vars, then all e vars
    s_off: Dict[int, int] = {}
    off = 0
    for i in owners:
        s_off[i] = off
        off += nstrat[i]
    n_svars = off

    # edges between *owner* nodes (follow nodes transfer specs unchanged):
    # resharding cost between producer root's out_spec and consumer root's
    # required in_spec for that input.
    edge_list: List[Tuple[int, int, np.ndarray]] = []  # (pi, ci, cost[r,c])
    for ci, n in enumerate(nodes):
        croot = owner_of[ci]
        for slot, pi in enumerate(n.inputs):
            proot = owner_of[pi]
            if croot == proot:
                continue  # follow edge: same strategy, no resharding
            ptensor = nodes[pi].out_bytes
            cost = np.zeros((nstrat[proot], nstrat[croot]))
            for r, ps in enumerate(nodes[proot].strategies):
                for c, cs in enumerate(nodes[croot].strategies):
                    # consumer's required spec for this input slot; a
                    # follow-node consumer is elementwise-aligned with its
                    # owner's output spec
                    if nodes[ci].follow is None and \
                            slot < len(cs.in_specs):
                        want = cs.in_specs[slot]
                    else:
                        want = cs.out_spec
                    cost[r, c] = edge_cost_factor * mesh.resharding_cost(
                        ptensor, ps.out_spec, want)
            edge_list.append((proot, croot, cost))

    e_off = []
    for (pi, ci, cost) in edge_list:
        e_off.append(off)
        off += cost.size
    n_vars = off

    # objective
    obj = np.zeros(n_vars)
    for i in owners:
        for r, st in enumerate(nodes[i].strategies):
            obj[s_off[i] + r] = st.compute_cost + st.comm_cost
    for k, (pi, ci, cost) in enumerate(edge_list):
        obj[e_off[k]:e_off[k] + cost.size] = cost.ravel()

    constraints = []
    # one-hot per node
    for i in owners:
        a = lil_matrix((1, n_vars))
        a[0, s_off[i]:s_off[i] + nstrat[i]] = 1.0
        constraints.append(LinearConstraint(a.tocsr(), 1.0, 1.0))
    # per-edge one-hot + consistency
    for k, (pi, ci, cost) in enumerate(edge_list):
        R, C = cost.shape
        a = lil_matrix((1, n_vars))
        a[0, e_off[k]:e_off[k] + R * C] = 1.0
        constraints.append(LinearConstraint(a.tocsr(), 1.0, 1.0))
        # sum_c e[r,c] <= s_i[r]
        a = lil_matrix((R, n_vars))
        for r in range(R):
            for c in range(C):
                a[r, e_off[k] + r * C + c] = 1.0
            a[r, s_off[pi] + r] = -1.0
        constraints.append(LinearConstraint(a.tocsr(), -np.inf, 0.0))
        a = lil_matrix((C, n_vars))
        for c in range(C):
            for r in range(R):
                a[c, e_off[k] + r * C + c] = 1.0
            a[c, s_off[ci] + c] = -1.0
        constraints.append(LinearConstraint(a.tocsr(), -np.inf, 0.0))
    # memory: follow nodes contribute their out_bytes at the owner's spec
    if memory_budget is not None:
        a = lil_matrix((1, n_vars))
        for i in owners:
            extra = sum(nodes[j].out_bytes for j in range(len(nodes))
                        if j != i and owner_of[j] == i)
            for r, st in enumerate(nodes[i].strategies):
                shard = mesh.shard_factor(st.out_spec)
                a[0, s_off[i] + r] = st.memory + extra / shard
        constraints.append(LinearConstraint(a.tocsr(), 0.0, memory_budget))

    res = milp(c=obj, constraints=constraints,
               integrality=np.ones(n_vars),
               bounds=Bounds(0.0, 1.0),
               options={"time_limit": time_limit})
    if res.x is None:
        return SolveResult(objective=float("inf"), choices={},
                           feasible=False)
    x = np.round(res.x).astype(int)
    choices = {}
    for i in owners:
        r = int(np.argmax(x[s_off[i]:s_off[i] + nstrat[i]]))
        choices[i] = r
    for j in range(len(nodes)):
        choices[j] = choices[owner_of[j]]
    return SolveResult(objective=float(res.fun), choices=choices,
                       feasible=True)
