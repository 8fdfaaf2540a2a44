"""Op-graph IR for the intra-operator auto-sharding solver.

The reference solves sharding over XLA HLO (C++ pass + ILP,
auto_sharding.py:617-872, strategy enumeration mirrored in
playground/auto_sharding_solver/hlo.py).  Our executor shards at the
module/op level, so the IR is an explicit op graph: nodes carry tensor
shapes + parameter sizes, strategies carry sharding specs and alpha-beta
costs, and the ILP picks one strategy per node plus resharding on edges.

Sharding spec: for each tensor dim, the logical-mesh axis it is split
along (or None).  Activations here are rank-2 [tokens, features]; a spec
is a 2-tuple, e.g. (0, None) = batch split along mesh axis 0 (DP),
(None, 1) = feature split along mesh axis 1 (TP).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

Spec = Tuple[Optional[int], Optional[int]]

REPLICATED: Spec = (None, None)


@dataclass
class Strategy:
    """One sharding choice for a node (reference: per-instruction strategy
    vector, hlo.py:664 build_strategy_and_cost)."""
    name: str
    in_specs: List[Spec]          # required spec per input edge
    out_spec: Spec
    compute_cost: float           # seconds (flops / device throughput)
    comm_cost: float              # seconds (internal collectives)
    memory: float                 # bytes per device (params + output act)


@dataclass
class Node:
    name: str
    op: str                       # "matmul" | "embedding" | "elemwise" | ...
    inputs: List[int] = field(default_factory=list)  # producer node ids
    out_bytes: float = 0.0        # full (unsharded) output activation bytes
    strategies: List[Strategy] = field(default_factory=list)
    #: follow: inherit the strategy index of this producer (s_follow,
    #: reference auto_sharding.py:716) — no ILP variable for this node.
    follow: Optional[int] = None


@dataclass
class Graph:
    nodes: List[Node] = field(default_factory=list)

    def add(self, node: Node) -> int:
        self.nodes.append(node)
        return len(self.nodes) - 1

    def edges(self) -> List[Tuple[int, int]]:
        out = []
        for i, n in enumerate(self.nodes):
            for p in n.inputs:
                out.append((p, i))
        return out


@dataclass
class MeshModel:
    """Alpha-beta cost model of a logical 2-D mesh over single-node xGMI
    (reference LogicalDeviceMesh, auto_sharding.py:81-141; on MI355X both
    mesh dims see the same per-link beta — full 7-link crossbar)."""
    shape: Tuple[int, int]
    alpha: float = 1e-5
    beta: float = 1.0 / 150e9      # s/byte per xGMI ring link

    def axis_size(self, axis: Optional[int]) -> int:
        return 1 if axis is None else self.shape[axis]

    def num_devices(self) -> int:
        return self.shape[0] * self.shape[1]

    # collective costs on one axis, B = full tensor bytes (per group)
    def all_gather(self, axis: int, bytes_: float) -> float:
        n = self.shape[axis]
        if n == 1:
            return 0.0
        return self.alpha + self.beta * bytes_ * (n - 1) / n

    def all_reduce(self, axis: int, bytes_: float) -> float:
        n = self.shape[axis]
        if n == 1:
            return 0.0
        return self.alpha + 2 * self.beta * bytes_ * (n - 1) / n

    def reduce_scatter(self, axis: int, bytes_: float) -> float:
        n = self.shape[axis]
        if n == 1:
            return 0.0
        return self.alpha + self.beta * bytes_ * (n - 1) / n

    def all_to_all(self, axis: int, bytes_: float) -> float:
        n = self.shape[axis]
        if n == 1:
            return 0.0
        # reference applies an n/2 congestion penalty (auto_sharding.py:136)
        return self.alpha + self.beta * bytes_ * (n - 1) / (n * n) * (n / 2)

    def shard_factor(self, spec: Spec) -> int:
        f = 1
        for ax in spec:
            if ax is not None:
                f *= self.shape[ax]
        return f

    def resharding_cost(self, bytes_full: float, src: Spec, dst: Spec
                        ) -> float:
        """Cost to convert a tensor from src spec to dst spec (reference
        cluster_env.py:92).  bytes_full = unsharded tensor bytes."""
        if src == dst:
            return 0.0
        cost = 0.0
        for d in range(2):
            ax = src[d]
            if ax is None or dst[d] == ax:
                continue
            if ax in dst:
                # same mesh axis splits a different tensor dim: all-to-all
                cost += self.all_to_all(ax, bytes_full)
            else:
                # axis dropped: all-gather over it
                cost += self.all_gather(ax, bytes_full)
        # replicated -> split transitions are local slices: free
        return cost


def dtype_bytes(dtype: str = "bf16") -> int:
    return {"bf16": 2, "fp16": 2, "fp32": 4}[dtype]
