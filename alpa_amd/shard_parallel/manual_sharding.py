"""Manual sharding: pjit-style per-module partition specs for PLAIN
torch models.

Capability analog of the reference's ``shard_parallel/manual_sharding.py``
(ManualShardingOption: user PartitionSpecs -> HLO shardings, solver
bypassed).  Here the spec maps module paths (fnmatch patterns) to a
partition kind and `apply_manual_sharding` REBUILDS each matched
``nn.Linear``/``nn.Embedding`` as the TP layer, copying this rank's
weight shard in — so any plain torch model becomes tensor-parallel
without touching its code:

    specs = {"blocks.*.mlp.fc1": "column", "blocks.*.mlp.fc2": "row",
             "embed": "vocab"}
    model = apply_manual_sharding(model, specs, mesh, axis=1)

Kinds: "column" (out-features sharded; bias sharded), "row" (in-features
sharded; bias replicated, output all-reduced), "vocab"
(nn.Embedding rows sharded with masked lookup + all-reduce).
"""
from __future__ import annotations

import fnmatch
from typing import Dict, Optional

import torch
import torch.nn as nn

from ..mesh import DeviceMesh
from ..parallel.layers import (ColumnParallelLinear, RowParallelLinear,
                               VocabParallelEmbedding)


def _shard_rows(w: torch.Tensor, tp: int, idx: int) -> torch.Tensor:
    per = w.shape[0] // tp
    return w[idx * per:(idx + 1) * per]


def _shard_cols(w: torch.Tensor, tp: int, idx: int) -> torch.Tensor:
    per = w.shape[1] // tp
    return w[:, idx * per:(idx + 1) * per]


def _convert(mod: nn.Module, kind: str, mesh: DeviceMesh, axis: int):
    tp = mesh.axis_size(axis)
    idx = max(mesh.axis_index(axis), 0) if mesh.is_member else 0
    dtype = next(mod.parameters()).dtype
    device = next(mod.parameters()).device
    with torch.no_grad():
        if kind == "column":
            assert isinstance(mod, nn.Linear), type(mod)
            assert mod.out_features % tp == 0
            new = ColumnParallelLinear(mod.in_features, mod.out_features,
                                       mesh, axis,
                                       bias=mod.bias is not None,
                                       dtype=dtype, device=device)
            new.weight.copy_(_shard_rows(mod.weight, tp, idx))
            if mod.bias is not None:
                new.bias.copy_(_shard_rows(mod.bias.unsqueeze(-1), tp,
                                           idx).squeeze(-1))
            return new
        if kind == "row":
            assert isinstance(mod, nn.Linear), type(mod)
            assert mod.in_features % tp == 0
            new = RowParallelLinear(mod.in_features, mod.out_features,
                                    mesh, axis,
                                    bias=mod.bias is not None,
                                    dtype=dtype, device=device)
            new.weight.copy_(_shard_cols(mod.weight, tp, idx))
            if mod.bias is not None:
                new.bias.copy_(mod.bias)
            return new
        if kind == "vocab":
            assert isinstance(mod, nn.Embedding), type(mod)
            assert mod.num_embeddings % tp == 0
            new = VocabParallelEmbedding(mod.num_embeddings,
                                         mod.embedding_dim, mesh, axis,
                                         dtype=dtype, device=device)
            new.weight.copy_(_shard_rows(mod.weight, tp, idx))
            return new
        if kind == "conv_column":
            from ..parallel.layers import ColumnParallelConv2d
            assert isinstance(mod, nn.Conv2d), type(mod)
            return ColumnParallelConv2d(mod, mesh, axis)
        if kind == "conv_row":
            from ..parallel.layers import RowParallelConv2d
            assert isinstance(mod, nn.Conv2d), type(mod)
            return RowParallelConv2d(mod, mesh, axis)
    raise ValueError(f"unknown partition kind {kind!r}")


def apply_manual_sharding(model: nn.Module, specs: Dict[str, str],
                          mesh: Optional[DeviceMesh],
                          axis: int = 1) -> nn.Module:
    """Replace every module whose path matches a spec pattern with its TP
    layer, weights sharded for THIS rank.  Returns the model (modified in
    place).  A no-op when mesh is None or the axis has size 1."""
    if mesh is None or mesh.axis_size(axis) == 1:
        return model
    matched = set()
    for name, mod in list(model.named_modules()):
        for pat, kind in specs.items():
            if fnmatch.fnmatch(name, pat):
                parent_path, _, leaf = name.rpartition(".")
                parent = model.get_submodule(parent_path) if parent_path \
                    else model
                setattr(parent, leaf, _convert(mod, kind, mesh, axis))
                matched.add(pat)
                break
    unmatched = set(specs) - matched
    assert not unmatched, f"specs matched no module: {sorted(unmatched)}"
    return model
