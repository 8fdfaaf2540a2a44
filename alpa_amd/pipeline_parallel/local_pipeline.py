"""Single-process sequential pipeline execution for debugging.

Capability analog of the reference's ``pipeline_parallel/local_pipeline.py``
(LocalPipelineRunner:16, compile_local_pipeline_executable): run every
stage of a PipelineModelSpec in ONE process, microbatch by microbatch, so
stage-splitting bugs can be separated from distributed-runtime bugs.  All
stages live in local memory; activations flow between them as ordinary
autograd tensors, so one ``loss.backward()`` differentiates the whole
chain — no p2p, no schedules.
"""
from __future__ import annotations

from typing import Any, List, Optional

import torch

from .layer_clustering import cluster_layers, uniform_layer_costs
from .spec import PipelineModelSpec


class LocalPipelineRunner:
    """Builds all P stages locally and runs (micro-)batches through them
    sequentially.  The result must equal both the serial model and the
    distributed PipelineEngine — the debugging midpoint between them."""

    def __init__(self, spec: PipelineModelSpec, num_stages: int,
                 dtype: torch.dtype = torch.float32, device=None,
                 layer_costs: Optional[List[float]] = None):
        P = num_stages
        costs = layer_costs or spec.layer_costs or \
            uniform_layer_costs(spec.num_layers)
        self.ranges = cluster_layers(costs, P)
        self.stages = torch.nn.ModuleList([
            spec.build_stage(layer_range=self.ranges[s],
                             is_first=(s == 0), is_last=(s == P - 1),
                             mesh=None, axis=1, dtype=dtype, device=device)
            for s in range(P)
        ])
        # tied groups within one process: alias the parameters directly
        # (same semantics as the cross-stage allreduce keeping them equal)
        for tg in (spec.tied_groups or []):
            items = sorted(((P + s) % P, path) for s, path in tg.items())
            s0, p0 = items[0]
            first = self._lookup(self.stages[s0], p0)
            for s, path in items[1:]:
                parent_path, _, leaf = path.rpartition(".")
                parent = self._lookup(self.stages[s], parent_path) \
                    if parent_path else self.stages[s]
                setattr(parent, leaf, first)

    @staticmethod
    def _lookup(module, path):
        obj = module
        for attr in path.split("."):
            obj = getattr(obj, attr)
        return obj

    def forward(self, microbatch: Any) -> torch.Tensor:
        """One microbatch through every stage; returns the loss."""
        x = None
        for s, stage in enumerate(self.stages):
            x = stage(x, microbatch)
        return x

    def train_step(self, microbatches: List[Any]) -> torch.Tensor:
        """Grad-accumulated step over microbatches; returns the mean loss.
        Caller owns the optimizer (grads are left on the parameters,
        scaled by 1/num_microbatches)."""
        total = None
        for mb in microbatches:
            loss = self.forward(mb) / len(microbatches)
            loss.backward()
            total = loss.detach() if total is None else total + loss.detach()
        return total

    def parameters(self):
        seen = set()
        for st in self.stages:
            for p in st.parameters():
                if id(p) not in seen:
                    seen.add(id(p))
                    yield p
