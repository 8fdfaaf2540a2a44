"""Graph capture: trace an arbitrary user ``nn.Module`` into the op-graph
IR the auto-sharding ILP solves over.

This is the capability the reference gets from jax tracing
(``alpa/shard_parallel/compile_executable.py:54`` traces the user's train
step to a jaxpr; ``alpa/util.py:868`` trace_jaxpr_with_micro_batch) — here
the program is a torch module, so capture = ``torch.fx.symbolic_trace`` +
shape propagation with the example batch.  The captured description is
mesh-INDEPENDENT (`OpDesc` records op kind, shapes and the owning module
path); per-mesh strategy enumeration happens in
``auto_sharding.build_captured_graph``.

Node kinds and how they map to ILP strategies:

  matmul     nn.Linear call — full dot strategy space (column / row /
             batch splits, strategies.matmul_strategies, mirroring the
             reference dot handler hlo.py:664-830)
  embedding  nn.Embedding — replicated or vocab-split table
  conv       nn.Conv2d — treated as a matmul over [N*H*W, Cin*k*k]
  norm       feature-normalizing ops (LayerNorm/softmax(dim=-1)/
             cross-entropy) — require a feature-REPLICATED input, so the
             ILP prices an all-gather on any feature-sharded producer
             edge (this is what makes "shard the MLP, replicate the
             attention" plans emerge for models whose attention is
             written with raw reshape/bmm)
  elemwise   true elementwise ops (gelu, add, mul, dropout...) — follow
             their producer's sharding (s_follow, auto_sharding.py:716)
  opaque     anything shape-changing or unrecognized (reshape mixing the
             feature dim, bmm, unknown modules) — requires replicated
             features, follows the batch split
  input      placeholders — arrive pre-sharded on the batch dim for free
  output     model outputs — must be feature-replicated (the user's loss
             runs outside the captured graph)
"""
from __future__ import annotations

import operator
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class OpDesc:
    """Mesh-independent description of one captured op."""
    name: str                      # fx node name
    kind: str                      # see module docstring
    inputs: List[int]              # producer OpDesc indices
    out_shape: Tuple[int, ...]
    module_path: Optional[str] = None
    #: matmul: {"k":, "n":}; embedding: {"vocab":, "h":}
    extra: Dict[str, Any] = field(default_factory=dict)


@dataclass
class CapturedGraph:
    ops: List[OpDesc]
    #: module_path -> OpDesc index (for plan application)
    by_module: Dict[str, int] = field(default_factory=dict)

    def matmul_like(self) -> List[OpDesc]:
        return [d for d in self.ops
                if d.kind in ("matmul", "embedding", "conv")]


# true elementwise functions: output spec == input spec, any sharding OK
_ELEMWISE_FNS = {
    F.gelu, F.relu, F.silu, F.sigmoid, F.tanh, F.dropout,
    torch.nn.functional.leaky_relu,
    torch.add, torch.sub, torch.mul, torch.div, torch.neg, torch.tanh,
    torch.sigmoid, torch.relu, torch.abs, torch.exp, torch.clamp,
    torch.where, torch.pow, torch.rsqrt, torch.sqrt,
    operator.add, operator.sub, operator.mul, operator.truediv,
    operator.neg, operator.pow,
}
_ELEMWISE_METHODS = {
    "add", "sub", "mul", "div", "neg", "tanh", "sigmoid", "relu", "abs",
    "exp", "clamp", "pow", "float", "to", "type_as", "contiguous",
    "clone", "detach", "masked_fill", "add_", "mul_",
}
_ELEMWISE_MODULES = (nn.GELU, nn.ReLU, nn.SiLU, nn.Sigmoid, nn.Tanh,
                     nn.Dropout, nn.Identity, nn.LeakyReLU,
                     # channel-preserving CNN ops: per-channel math, so
                     # they FOLLOW a channel-sharded conv producer
                     nn.BatchNorm2d, nn.MaxPool2d, nn.AvgPool2d,
                     nn.AdaptiveAvgPool2d)
# feature-normalizing: need the full feature dim present
_NORM_FNS = {F.layer_norm, F.softmax, F.log_softmax, F.cross_entropy,
             F.rms_norm} if hasattr(F, "rms_norm") else {
    F.layer_norm, F.softmax, F.log_softmax, F.cross_entropy}
_NORM_MODULES = (nn.LayerNorm, nn.Softmax, nn.CrossEntropyLoss)
# batch-shape-preserving views: follow (feature dim untouched) when the
# last dim is unchanged
_VIEW_FNS = {torch.reshape, torch.flatten, torch.unsqueeze, torch.squeeze}
_VIEW_METHODS = {"view", "reshape", "flatten", "unsqueeze", "squeeze",
                 "expand", "permute", "transpose"}


def _shape_of(node) -> Optional[Tuple[int, ...]]:
    tm = node.meta.get("tensor_meta")
    if tm is None:
        return None
    if hasattr(tm, "shape"):
        return tuple(tm.shape)
    return None


def _tensor_args(node) -> List["torch.fx.Node"]:
    out = []

    def visit(a):
        import torch.fx
        if isinstance(a, torch.fx.Node):
            out.append(a)
        elif isinstance(a, (list, tuple)):
            for x in a:
                visit(x)
        elif isinstance(a, dict):
            for x in a.values():
                visit(x)

    for a in node.args:
        visit(a)
    for a in node.kwargs.values():
        visit(a)
    return out


def capture_graph(model: nn.Module, example_inputs) -> CapturedGraph:
    """Trace `model` with the example batch and classify every fx node.

    `example_inputs`: tuple of positional tensors for model.forward.
    """
    from torch.fx import symbolic_trace
    from torch.fx.passes.shape_prop import ShapeProp

    gm = symbolic_trace(model)
    if not isinstance(example_inputs, (list, tuple)):
        example_inputs = (example_inputs,)
    # shape propagation EXECUTES the graph; run it in eval + no_grad so
    # stateful modules (BatchNorm running stats) are not mutated by the
    # capture pass (gm shares buffers with the user's model)
    was_training = gm.training
    gm.eval()
    with torch.no_grad():
        ShapeProp(gm).propagate(*example_inputs)
    if was_training:
        gm.train()

    ops: List[OpDesc] = []
    idx_of: Dict[str, int] = {}
    cap = CapturedGraph(ops)

    def add(desc: OpDesc, fx_name: str) -> int:
        ops.append(desc)
        idx_of[fx_name] = len(ops) - 1
        return len(ops) - 1

    def producer_ids(node) -> List[int]:
        out = []
        for a in _tensor_args(node):
            if a.name in idx_of:
                out.append(idx_of[a.name])
        return out

    for node in gm.graph.nodes:
        shape = _shape_of(node)
        if node.op == "placeholder":
            if shape is not None:
                add(OpDesc(node.name, "input", [], shape), node.name)
            continue
        if node.op == "get_attr":
            continue  # parameters reached via modules; raw attrs are
            #           treated as constants (their consumers go opaque)
        if node.op == "output":
            prods = producer_ids(node)
            add(OpDesc(node.name, "output", prods, shape or ()), node.name)
            continue

        prods = producer_ids(node)
        if shape is None:
            # non-tensor result (e.g. size()); ignore
            continue

        if node.op == "call_module":
            mod = gm.get_submodule(node.target)
            path = str(node.target)
            if isinstance(mod, nn.Linear):
                i = add(OpDesc(node.name, "matmul", prods[:1], shape,
                               module_path=path,
                               extra={"k": mod.in_features,
                                      "n": mod.out_features}), node.name)
                cap.by_module[path] = i
            elif isinstance(mod, nn.Embedding):
                i = add(OpDesc(node.name, "embedding", [], shape,
                               module_path=path,
                               extra={"vocab": mod.num_embeddings,
                                      "h": mod.embedding_dim}), node.name)
                cap.by_module[path] = i
            elif isinstance(mod, nn.Conv2d):
                k = mod.in_channels * mod.kernel_size[0] * \
                    mod.kernel_size[1] // mod.groups
                i = add(OpDesc(node.name, "conv", prods[:1], shape,
                               module_path=path,
                               extra={"k": k, "n": mod.out_channels,
                                      "cin": mod.in_channels,
                                      "groups": mod.groups}),
                        node.name)
                cap.by_module[path] = i
            elif isinstance(mod, _NORM_MODULES):
                add(OpDesc(node.name, "norm", prods[:1], shape,
                           module_path=path), node.name)
            elif isinstance(mod, _ELEMWISE_MODULES):
                add(OpDesc(node.name, "elemwise", prods[:1], shape,
                           module_path=path), node.name)
            else:
                # unknown module: conservatively needs replicated features
                add(OpDesc(node.name, "opaque", prods, shape,
                           module_path=path), node.name)
        elif node.op == "call_function":
            fn = node.target
            if getattr(fn, "__name__", "") == "_boundary_identity":
                # pipeline marker: sharding-transparent identity
                add(OpDesc(node.name, "elemwise", prods[:1], shape),
                    node.name)
            elif fn in _NORM_FNS:
                add(OpDesc(node.name, "norm", prods[:1], shape), node.name)
            elif fn in _ELEMWISE_FNS:
                add(OpDesc(node.name, "elemwise", prods, shape), node.name)
            elif fn is operator.getitem:
                add(OpDesc(node.name, "elemwise", prods[:1], shape),
                    node.name)
            elif fn in _VIEW_FNS or fn in (torch.permute, torch.transpose):
                kind = _view_kind(node, shape)
                add(OpDesc(node.name, kind, prods[:1], shape), node.name)
            elif fn in (torch.matmul, torch.bmm, torch.einsum):
                # activation-activation contraction (attention scores):
                # executable only on replicated features
                add(OpDesc(node.name, "opaque", prods, shape), node.name)
            else:
                add(OpDesc(node.name, "opaque", prods, shape), node.name)
        elif node.op == "call_method":
            m = node.target
            if m in _ELEMWISE_METHODS:
                add(OpDesc(node.name, "elemwise", prods[:1], shape),
                    node.name)
            elif m in _VIEW_METHODS:
                add(OpDesc(node.name, _view_kind(node, shape), prods[:1],
                           shape), node.name)
            elif m in ("softmax", "log_softmax"):
                add(OpDesc(node.name, "norm", prods[:1], shape), node.name)
            elif m in ("sum", "mean", "max", "min"):
                add(OpDesc(node.name, "elemwise", prods[:1], shape),
                    node.name)
            else:
                add(OpDesc(node.name, "opaque", prods, shape), node.name)
    return cap


def _view_kind(node, out_shape) -> str:
    """A view that PRESERVES the trailing (feature) dim only reshapes the
    batch dims -> sharding follows; anything touching the feature dim is
    opaque (reference reshape/transpose follow rules, hlo.py:87)."""
    ins = _tensor_args(node)
    if not ins:
        return "opaque"
    in_shape = _shape_of(ins[0])
    if in_shape and out_shape and len(in_shape) >= 1 and \
            len(out_shape) >= 1 and in_shape[-1] == out_shape[-1]:
        return "elemwise"
    return "opaque"
