"""Per-op sharding strategy enumeration over a 2-D logical mesh.

Semantics mirror the reference's dot handler (C++
auto_sharding_dot_handler.cc; readable Python mirror
``playground/auto_sharding_solver/hlo.py:664-830``): batch-dim splits
(SR = SS x SR), column parallel (RS = RR x RS), row parallel with
all-reduce (RR = RS x SR + AR), 2-D combinations, replicated — each with
alpha-beta communication cost, per-device compute cost, and per-device
memory.  Costs are in seconds; memory in bytes.
"""
from __future__ import annotations

from typing import List, Optional

import os

from .ir import MeshModel, Node, Spec, Strategy, REPLICATED

# effective per-device matmul throughput used to convert FLOPs -> seconds;
# fallback when no profiled database exists (measured hipBLASLt bf16 rate
# on MI355X, profiles/)
MATMUL_TFLOPS = 1.2e15

_CALIBRATED: float = None


def effective_matmul_flops() -> float:
    """Measured matmul rate from the profiled cost DB when present
    (tools/calibrate.py -> prof_database.pkl), else the fallback constant —
    the "cost model recalibrated from measured counters" loop."""
    global _CALIBRATED
    if _CALIBRATED is None:
        _CALIBRATED = MATMUL_TFLOPS
        from ..global_env import global_config
        path = global_config.prof_database_path
        if os.path.exists(path):
            try:
                from ..mesh_profiling import ProfilingResultDatabase
                db = ProfilingResultDatabase()
                db.load(path)
                for (key, shape), r in db.data.items():
                    if "matmul_bf16" in r.op_curves:
                        c = r.op_curves["matmul_bf16"]
                        _CALIBRATED = c.sizes[-1] / c.times[-1]
                        break
            except Exception:
                pass
    return _CALIBRATED
# grad multiplier: fwd + bwd matmuls (dX, dW)
TRAIN_FACTOR = 3.0
# Gradient all-reduce is overlapped with backward compute by the bucketed
# comm-stream synchronizer (parallel/grad_sync.py); only this fraction is
# exposed on the critical path.  TP activation all-reduces are synchronous.
GRAD_SYNC_OVERLAP = 0.25


def matmul_strategies(mesh: MeshModel, tokens: int, k: int, n: int,
                      dtype_bytes: int = 2,
                      train: bool = True) -> List[Strategy]:
    """Y[tokens, n] = X[tokens, k] @ W[k, n].  One input edge (X)."""
    flops = 2.0 * tokens * k * n * (TRAIN_FACTOR if train else 1.0)
    w_bytes = float(k) * n * dtype_bytes
    y_bytes = float(tokens) * n * dtype_bytes
    # training state per weight element: bf16 w + bf16 grad + fp32 m + fp32 v
    # = 12 bytes/param
    w_state = 12.0 * k * n if train else w_bytes

    out: List[Strategy] = []

    def dev(ax0: Optional[int], ax1: Optional[int]) -> int:
        return mesh.axis_size(ax0) * mesh.axis_size(ax1)

    axes = [None] + [a for a in (0, 1) if mesh.shape[a] > 1]
    for b_ax in axes:                   # batch (tokens) split axis
        for w_ax in axes:               # weight n-dim (column) split axis
            if b_ax is not None and b_ax == w_ax:
                continue
            d = dev(b_ax, w_ax)
            # batch split => dW = X^T dY is a partial sum over b_ax:
            # gradient all-reduce of the weight shard (this is where the
            # DP grad-sync cost enters the ILP, as in GSPMD)
            gsync = 0.0
            if train and b_ax is not None:
                gsync = GRAD_SYNC_OVERLAP * mesh.all_reduce(
                    b_ax, w_bytes / mesh.axis_size(w_ax))
            # column-split: backward dX = dY @ W^T sums partials over w_ax
            # (the _CopyToParallel bwd all-reduce, parallel/layers.py)
            dx_ar = 0.0
            if train and w_ax is not None:
                x_bytes = float(tokens) * k * dtype_bytes
                dx_ar = mesh.all_reduce(
                    w_ax, x_bytes / mesh.axis_size(b_ax))
            out.append(Strategy(
                name=f"b{b_ax}_col{w_ax}",
                in_specs=[(b_ax, None)],
                out_spec=(b_ax, w_ax),
                compute_cost=flops / d / effective_matmul_flops(),
                comm_cost=gsync + dx_ar,
                memory=w_state / mesh.axis_size(w_ax) + y_bytes / d))
        # row-parallel: X feature-split on ax, W k-split on ax, out
        # all-reduced over ax (partial sums)
        for r_ax in (0, 1):
            if r_ax == b_ax or mesh.axis_size(r_ax) == 1:
                continue
            d = dev(b_ax, r_ax)
            ar_bytes = y_bytes / mesh.axis_size(b_ax)
            # fwd output all-reduce; bwd dX = dY @ W^T is local (dY
            # replicated across r_ax) — no bwd comm
            cost_ar = mesh.all_reduce(r_ax, ar_bytes)
            gsync = 0.0
            if train and b_ax is not None:
                gsync = GRAD_SYNC_OVERLAP * mesh.all_reduce(
                    b_ax, w_bytes / mesh.axis_size(r_ax))
            out.append(Strategy(
                name=f"b{b_ax}_row{r_ax}",
                in_specs=[(b_ax, r_ax)],
                out_spec=(b_ax, None),
                compute_cost=flops / d / effective_matmul_flops(),
                comm_cost=cost_ar + gsync,
                memory=w_state / mesh.axis_size(r_ax) +
                y_bytes / dev(b_ax, None)))
    return out


def embedding_strategies(mesh: MeshModel, tokens: int, vocab: int, h: int,
                         dtype_bytes: int = 2) -> List[Strategy]:
    """Y[tokens, h] = gather(table[vocab, h], ids).  No tensor input."""
    w_bytes = float(vocab) * h * dtype_bytes
    w_state = 12.0 * vocab * h
    y_bytes = float(tokens) * h * dtype_bytes
    out = []
    axes = [None] + [a for a in (0, 1) if mesh.shape[a] > 1]
    for b_ax in axes:
        d = mesh.axis_size(b_ax)
        gsync = GRAD_SYNC_OVERLAP * mesh.all_reduce(b_ax, w_bytes) \
            if b_ax is not None else 0.0
        out.append(Strategy(
            name=f"b{b_ax}_repl", in_specs=[], out_spec=(b_ax, None),
            compute_cost=0.0, comm_cost=gsync,
            memory=w_state + y_bytes / d))
        for v_ax in (0, 1):
            if v_ax == b_ax or mesh.axis_size(v_ax) == 1:
                continue
            # vocab-split table: out needs all-reduce over v_ax
            ar = mesh.all_reduce(v_ax, y_bytes / d) * 2.0
            gsync = GRAD_SYNC_OVERLAP * \
                mesh.all_reduce(b_ax, w_bytes / mesh.axis_size(v_ax)) \
                if b_ax is not None else 0.0
            out.append(Strategy(
                name=f"b{b_ax}_vocab{v_ax}", in_specs=[],
                out_spec=(b_ax, None), compute_cost=0.0,
                comm_cost=ar + gsync,
                memory=w_state / mesh.axis_size(v_ax) + y_bytes / d))
    return out


def elemwise_follow_node(name: str, producer: int, out_bytes: float) -> Node:
    """Elementwise/norm/attention op following its producer's sharding
    (reference s_follow aliasing, auto_sharding.py:716)."""
    return Node(name=name, op="elemwise", inputs=[producer],
                out_bytes=out_bytes, follow=producer)


def loss_strategies(mesh: MeshModel, tokens: int, vocab: int,
                    dtype_bytes: int = 2) -> List[Strategy]:
    """Softmax-cross-entropy over logits [tokens, vocab]: batch split free;
    vocab split costs 3 small all-reduces [tokens] fp32 (vocab-parallel CE,
    parallel/layers.py)."""
    out = []
    axes = [None] + [a for a in (0, 1) if mesh.shape[a] > 1]
    for b_ax in axes:
        d = mesh.axis_size(b_ax)
        out.append(Strategy(name=f"b{b_ax}", in_specs=[(b_ax, None)],
                            out_spec=(b_ax, None), compute_cost=0.0,
                            comm_cost=0.0, memory=0.0))
        for v_ax in (0, 1):
            if v_ax == b_ax or mesh.axis_size(v_ax) == 1:
                continue
            small = 3 * mesh.all_reduce(v_ax, tokens / d * 4.0)
            out.append(Strategy(name=f"b{b_ax}_v{v_ax}",
                                in_specs=[(b_ax, v_ax)],
                                out_spec=(b_ax, None), compute_cost=0.0,
                                comm_cost=small, memory=0.0))
    return out


def input_strategies(mesh: MeshModel, out_bytes: float) -> List[Strategy]:
    """Graph inputs (placeholders): arrive pre-sharded on the batch dim
    for free (the SPMD data loader feeds per-rank shards)."""
    out = [Strategy(name="repl", in_specs=[], out_spec=REPLICATED,
                    compute_cost=0.0, comm_cost=0.0, memory=out_bytes)]
    for b_ax in (0, 1):
        if mesh.axis_size(b_ax) > 1:
            out.append(Strategy(
                name=f"b{b_ax}", in_specs=[], out_spec=(b_ax, None),
                compute_cost=0.0, comm_cost=0.0,
                memory=out_bytes / mesh.axis_size(b_ax)))
    return out


def replicated_feature_strategies(mesh: MeshModel, out_bytes: float,
                                  n_inputs: int = 1,
                                  name: str = "norm") -> List[Strategy]:
    """Ops that need the FULL feature dim locally (LayerNorm, softmax
    over features, raw bmm/reshape mixing features, unknown modules):
    batch split passes through, feature split must be gathered first —
    the resharding cost lands on the incoming edge."""
    out = []
    axes = [None] + [a for a in (0, 1) if mesh.shape[a] > 1]
    for b_ax in axes:
        d = mesh.axis_size(b_ax)
        out.append(Strategy(
            name=f"b{b_ax}", in_specs=[(b_ax, None)] * n_inputs,
            out_spec=(b_ax, None), compute_cost=0.0, comm_cost=0.0,
            memory=out_bytes / d))
    return out
