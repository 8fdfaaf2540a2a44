"""User-facing pipeline boundary markers + remat API for PLAIN torch
models (reference ``alpa/pipeline_parallel/primitive_def.py:18``
``mark_pipeline_boundary`` and ``layer_construction.py:542,571``
``manual_remat``/``automatic_remat``).

The reference inserts an identity custom-call into the traced program
that survives compilation; here the marker is an fx-visible identity
(`torch.fx.wrap`), so tracing a plain model that calls
``mark_pipeline_boundary(x)`` yields a graph that ``spec_from_module``
can slice into pipeline stages — the module-level analog of
``slice_closed_jaxpr_by_full_pipeline_marks`` (computation.py:387).

    class Net(nn.Module):
        def forward(self, x):
            x = self.part1(x)
            x = alpa_amd.mark_pipeline_boundary(x)
            return self.part2(x)

    spec = spec_from_module(lambda: Net(), example_batch,
                            loss_fn=lambda out, mb: F.mse_loss(out, mb["y"]))
    state = TrainState.create(spec, PipeshardParallel(...))

Remat:
    manual_remat(module)            wrap ONE module in activation remat
    automatic_remat(model, n)       slice the model's top-level children
                                    into n cost-balanced segments and
                                    remat each (layer-boundary remat,
                                    remat_sliced_eqns)
"""
from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional, Tuple

import torch
import torch.fx
import torch.nn as nn

from .spec import PipelineModelSpec


def _boundary_identity(x):
    return x


def mark_pipeline_boundary(x: torch.Tensor) -> torch.Tensor:
    """Identity marking a pipeline stage boundary inside ``forward``
    (reference primitive_def.py:18).  Under fx tracing it inserts an
    explicit graph node (proxy-aware, so ``aa.mark_pipeline_boundary``
    works however it is imported); a no-op when the model runs
    un-pipelined."""
    if isinstance(x, torch.fx.Proxy):
        return x.tracer.create_proxy("call_function", _boundary_identity,
                                     (x,), {})
    return x


def _split_at_boundaries(model: nn.Module):
    """Trace and split the model at mark_pipeline_boundary calls.
    Returns (segments, gm): segments = list of fx GraphModules, where
    segment i's forward takes (the model inputs for i == 0, else the
    boundary tensor) and returns the next boundary tensor (or the model
    output for the last)."""
    from torch.fx.passes.split_module import split_module
    gm = torch.fx.symbolic_trace(model)
    stage_of: Dict[torch.fx.Node, int] = {}
    cur = 0
    for node in gm.graph.nodes:
        if node.op == "call_function" and \
                getattr(node.target, "__name__", "") == \
                "_boundary_identity":
            stage_of[node] = cur
            cur += 1
        else:
            stage_of[node] = cur
    n_stages = cur + 1
    if n_stages == 1:
        return [gm], gm
    split = split_module(gm, model, lambda n: stage_of[n])
    segments = [getattr(split, f"submod_{i}") for i in range(n_stages)]
    return segments, gm


def spec_from_module(builder: Callable[[], nn.Module], example_batch: Any,
                     loss_fn: Callable[[Any, Any], torch.Tensor],
                     batch_to_inputs: Optional[Callable] = None,
                     layer_costs: Optional[List[float]] = None
                     ) -> PipelineModelSpec:
    """Build a PipelineModelSpec from a PLAIN torch model with
    ``mark_pipeline_boundary`` calls in its forward.

    - ``builder()`` constructs the model (seed inside, so every rank
      builds identical weights).
    - ``example_batch``: one microbatch dict/tuple; ``batch_to_inputs``
      maps it to the model's positional inputs (default: ``mb["x"]``).
    - ``loss_fn(output, microbatch)`` computes the scalar loss on the
      LAST stage.

    The marked segments become the clustering layers (one layer per
    segment); the stage builder re-traces on each rank and keeps only
    its layer range, chaining segment forwards.
    """
    to_inputs = batch_to_inputs or (lambda mb: (mb["x"],))
    probe = builder()
    segments, _ = _split_at_boundaries(probe)
    n_seg = len(segments)
    with torch.no_grad():
        x = to_inputs(example_batch)
        act = segments[0](*x)
        act_shape = tuple(act.shape)
    if layer_costs is None:
        layer_costs = [
            max(sum(p.numel() for p in seg.parameters()), 1.0)
            for seg in segments
        ]

    def build_stage(layer_range, is_first, is_last, mesh, axis, dtype,
                    device):
        lo, hi = layer_range
        model = builder().to(device=device or "cpu", dtype=dtype)
        segs, _ = _split_at_boundaries(model)
        mine = segs[lo:hi]

        class Stage(nn.Module):
            def __init__(self):
                super().__init__()
                self.segs = nn.ModuleList(mine)

            def forward(self, x, microbatch):
                if is_first:
                    y = self.segs[0](*to_inputs(microbatch))
                    rest = list(self.segs)[1:]
                else:
                    y = x
                    rest = list(self.segs)
                for seg in rest:
                    y = seg(y)
                if is_last:
                    return loss_fn(y, microbatch)
                return y

        return Stage()

    def act_shape_fn(mb):
        b = to_inputs(mb)[0].shape[0]
        return (b,) + tuple(act_shape[1:])

    return PipelineModelSpec(num_layers=n_seg, build_stage=build_stage,
                             act_shape=act_shape_fn,
                             layer_costs=list(layer_costs))


# ----------------------------------------------------------------------
# Remat (reference manual_remat / automatic_remat,
# layer_construction.py:542,571)
# ----------------------------------------------------------------------


class _Remat(nn.Module):
    """Activation-rematerializing wrapper: forward under
    torch.utils.checkpoint (non-reentrant), so only the module's inputs
    survive to backward and its internals recompute."""

    def __init__(self, inner: nn.Module):
        super().__init__()
        self.inner = inner

    def forward(self, *args, **kwargs):
        if torch.is_grad_enabled() and any(
                torch.is_tensor(a) and a.requires_grad for a in args):
            from torch.utils.checkpoint import checkpoint
            return checkpoint(self.inner, *args, use_reentrant=False,
                              **kwargs)
        return self.inner(*args, **kwargs)


def manual_remat(module: nn.Module) -> nn.Module:
    """Wrap one module in activation remat (reference manual_remat)."""
    return _Remat(module)


def automatic_remat(model: nn.Module, num_layers: int = 0) -> nn.Module:
    """Slice the model's layer sequence into cost-balanced segments and
    remat each (reference automatic_remat: remat at auto-clustered layer
    boundaries).  Operates on the model's ModuleList children (the
    clustering unit of the module-level world); num_layers = 0 remats
    every block individually.

    Grouping contract: when num_layers < len(blocks), the model's
    forward must iterate the list as ``for b in blocks: x = b(x)``
    (self-contained blocks) — grouped blocks are chained inside one
    nn.Sequential, so any per-block work done OUTSIDE the block in
    forward would be applied per GROUP instead."""
    from .layer_clustering import cluster_layers
    for name, child in model.named_children():
        if isinstance(child, nn.ModuleList) and len(child) > 1:
            blocks = list(child)
            if num_layers and num_layers < len(blocks):
                costs = [max(sum(p.numel() for p in b.parameters()), 1.0)
                         for b in blocks]
                ranges = cluster_layers(costs, num_layers)
                grouped = []
                for (a, b) in ranges:
                    seq = blocks[a] if b - a == 1 else nn.Sequential(
                        *blocks[a:b])
                    grouped.append(_Remat(seq))
                setattr(model, name, nn.ModuleList(grouped))
            else:
                setattr(model, name, nn.ModuleList(
                    [_Remat(b) for b in blocks]))
            return model
    raise ValueError("automatic_remat: no ModuleList of blocks found; "
                     "use manual_remat on specific modules instead")
