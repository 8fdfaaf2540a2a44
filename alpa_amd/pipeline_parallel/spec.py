"""Pipeline model specification: how a model family exposes its layer
structure to the pipeline compiler.

The reference slices a traced jaxpr at pipeline markers
(``computation.py:387`` slice_closed_jaxpr_by_full_pipeline_marks); in the
module-level world the model family provides a stage builder and the
compiler decides the layer->stage assignment (auto clustering DP or
manual).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Callable, List, Optional, Sequence, Tuple

import torch


@dataclass
class PipelineModelSpec:
    """Everything the pipeline compiler needs from a model family.

    - ``num_layers``: number of cluster-able layers (transformer blocks).
    - ``build_stage(layer_range, is_first, is_last, mesh, axis, dtype,
      device) -> nn.Module``: builds this rank's stage module.  The stage
      forward signature is ``forward(x, microbatch)`` where x is None on
      the first stage; the last stage returns the microbatch loss.
    - ``act_shape(microbatch) -> tuple``: shape of the activation crossing
      stage boundaries for one microbatch.
    - ``layer_costs``: optional per-layer FLOPs estimate for the
      clustering DP (uniform if None).
    """
    num_layers: int
    build_stage: Callable[..., torch.nn.Module]
    act_shape: Callable[[Any], Tuple[int, ...]]
    layer_costs: Optional[Sequence[float]] = None
    #: tied parameters across stages: each dict maps stage index (-1 =
    #: last) -> param path in that stage's module; grads are all-reduced
    #: across the same-coordinate ranks of those stages each step
    #: (reference cross-mesh allreduce for tied embeddings, N15)
    tied_groups: Optional[list] = None
    #: optional per-MICROBATCH numbers for the profile-guided stage
    #: search (stage_construction.profiled_stage_search): fwd+bwd flops
    #: per layer, boundary activation bytes, bf16 param bytes per layer
    layer_flops: Optional[Sequence[float]] = None
    boundary_act_bytes: float = 0.0
    layer_param_bytes: Optional[Sequence[float]] = None
    #: optional MEASURED stage-cost curve name in the profiling DB
    #: (tools/profile_stages.py, e.g. "gpt_stage_cost_h2560") and this
    #: job's tokens per microbatch — the training DP then uses measured
    #: per-stage times instead of the flops/curve model
    stage_cost_curve: Optional[str] = None
    microbatch_tokens: Optional[float] = None
    #: optional activation layout per stage for the boundary exchange:
    #: fn(stage_idx, (dp, tp), act_rank) -> dim-partition tuple (None =
    #: the default batch-over-dp layout).  Declaring a feature-sharded
    #: boundary makes the engine reshard tiles instead of replicating
    #: (runtime.PipelineEngine.boundary_parts_fn)
    boundary_parts: Optional[Callable] = None
