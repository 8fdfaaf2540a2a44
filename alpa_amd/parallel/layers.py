"""Tensor-parallel execution primitives over a DeviceMesh axis.

These are the *execution substrate* the auto-sharding plan maps onto: the ILP
solver (shard_parallel/) picks per-matmul strategies (column-split /
row-split / replicated / batch-split, mirroring the strategy space of the
reference's C++ dot handler — see ``playground/auto_sharding_solver/hlo.py:664``
in /root/reference), and the planner instantiates these modules accordingly.

Collectives are RCCL over xGMI via the mesh axis group.  Autograd handles the
backward collectives through the `_CopyToParallel`/`_ReduceFromParallel`
function pair (an identity-fwd/allreduce-bwd and allreduce-fwd/identity-bwd
dual, the standard conjugate pair of column/row sharding).
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from .. import ops
from ..mesh import DeviceMesh, is_distributed


class _CopyToParallel(torch.autograd.Function):
    """Identity forward; all-reduce gradient backward (input of col-split)."""

    @staticmethod
    def forward(ctx, x, mesh: DeviceMesh, axis: int):
        ctx.mesh, ctx.axis = mesh, axis
        return x

    @staticmethod
    def backward(ctx, g):
        g = g.contiguous()
        ctx.mesh.all_reduce(g, axis=ctx.axis)
        return g, None, None


class _ReduceFromParallel(torch.autograd.Function):
    """All-reduce forward (output of row-split); identity backward."""

    @staticmethod
    def forward(ctx, x, mesh: DeviceMesh, axis: int):
        x = x.contiguous()
        mesh.all_reduce(x, axis=axis)
        return x

    @staticmethod
    def backward(ctx, g):
        return g, None, None


def copy_to_tp(x, mesh, axis):
    if mesh is None or mesh.axis_size(axis) == 1:
        return x
    return _CopyToParallel.apply(x, mesh, axis)


def reduce_from_tp(x, mesh, axis):
    if mesh is None or mesh.axis_size(axis) == 1:
        return x
    return _ReduceFromParallel.apply(x, mesh, axis)


def tag_seed(init_seed: int, tag: str) -> int:
    """Deterministic, position-independent seed for a named parameter:
    lets pipeline stages built on different ranks draw identical weights
    for the same logical layer (pipeline-vs-serial oracle tests)."""
    import zlib
    return (zlib.crc32(tag.encode()) ^ (init_seed * 0x9E3779B1)) % (2**31 - 1)


@torch.no_grad()
def _sharded_normal_(w: torch.Tensor, full_shape, shard_dim: int,
                     shard_idx: int, num_shards: int, std: float,
                     seed: int = None):
    """Initialize `w` as shard `shard_idx` (along `shard_dim`) of a full
    tensor drawn from N(0, std).

    With seed=None the seed is drawn from the *global* RNG — consumed
    identically on every rank since all ranks build the same module
    structure — so the union of shards equals the serial init exactly
    (serial-vs-parallel tests depend on this; cf. the reference's oracle
    pattern, alpa/testing.py:233).  Pipeline stages pass an explicit
    tag_seed instead (position-independent).
    """
    if seed is None:
        seed = int(torch.randint(0, 2**31 - 1, (1,)).item())
    gen_device = w.device if w.device.type == "cuda" else "cpu"
    g = torch.Generator(device=gen_device)
    g.manual_seed(seed)
    if num_shards == 1:
        w.normal_(0.0, std, generator=g)
        return
    full = torch.empty(full_shape, dtype=torch.float32, device=gen_device)
    full.normal_(0.0, std, generator=g)
    n = full_shape[shard_dim] // num_shards
    w.copy_(full.narrow(shard_dim, shard_idx * n, n))
    del full


def _fp8_ok(x, weight, module=None) -> bool:
    """Gate for the fp8 GEMM path (ops/fp8.py): opt-in via
    global_config.fp8_gemm, CUDA only, dims 16-aligned.  Modules marked
    ``_fp8_exclude`` (the LM head — standard mixed-fp8 practice keeps
    first/last layers in high precision; measured to close most of the
    fp8 loss-curve gap) stay bf16."""
    from ..global_env import global_config
    if not global_config.fp8_gemm or not x.is_cuda:
        return False
    if module is not None and getattr(module, "_fp8_exclude", False):
        return False
    if not torch.is_grad_enabled():
        # inference: hipBLASLt fp8 only beats bf16 from M>=128 (measured
        # tools/fp8_decode_probe.py: 0.92x at M<=64, 1.15x at 128, 1.81x
        # at 256) — so PREFILL goes fp8, per-token decode stays on the
        # skinny bf16 path (also skips the per-linear quantize launches)
        if x.numel() // x.shape[-1] < 128:
            return False
    from ..ops.fp8 import fp8_available
    return (fp8_available(x) and x.shape[-1] % 16 == 0 and
            weight.shape[0] % 16 == 0)


class ColumnParallelLinear(nn.Module):
    """Y = X @ W^T + b with W row-sharded over out_features (each rank holds
    out_features/tp rows).  Output stays sharded along the feature dim.

    GEMM goes to hipBLASLt via torch.matmul; `gelu=True` fuses the bias+GeLU
    epilogue into the hand-written HIP kernel (ops.bias_gelu).
    """

    def __init__(self, in_features: int, out_features: int,
                 mesh: Optional[DeviceMesh] = None, axis: int = 1,
                 bias: bool = True, gelu: bool = False,
                 dtype=torch.float32, device=None, init_seed=None,
                 init_tag: str = None, init_std: float = None):
        super().__init__()
        self.mesh, self.axis = mesh, axis
        tp = mesh.axis_size(axis) if mesh is not None else 1
        assert out_features % tp == 0, (out_features, tp)
        self.in_features = in_features
        self.out_features = out_features
        self.out_per_rank = out_features // tp
        self.gelu = gelu
        self.weight = nn.Parameter(
            torch.empty(self.out_per_rank, in_features, dtype=dtype,
                        device=device))
        idx = mesh.axis_index(axis) if (mesh is not None and mesh.is_member) else 0
        seed = tag_seed(init_seed, init_tag) if init_tag is not None else None
        std = init_std if init_std is not None else 1.0 / math.sqrt(in_features)
        _sharded_normal_(self.weight, (out_features, in_features), 0,
                         max(idx, 0), tp, std, seed)
        if bias:
            self.bias = nn.Parameter(
                torch.zeros(self.out_per_rank, dtype=dtype, device=device))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        x = copy_to_tp(x, self.mesh, self.axis)
        if ops.skinny_ok(x, self.weight, self):
            # decode-shaped inference (M <= 64): packed GEMV kernel
            if self.gelu:
                return ops.bias_gelu(
                    ops.skinny_linear(x, self.weight, None, self),
                    self.bias)
            return ops.skinny_linear(x, self.weight, self.bias, self)
        if self._use_fp8(x):
            from ..ops.fp8 import fp8_linear
            if self.gelu:
                return ops.bias_gelu(fp8_linear(x, self.weight,
                                                module=self), self.bias)
            return fp8_linear(x, self.weight, self.bias, module=self)
        if self.gelu:
            assert self.bias is not None
            return ops.bias_gelu(torch.matmul(x, self.weight.t()),
                                 self.bias)
        if self.bias is not None:
            # bias fused into the hipBLASLt epilogue (addmm)
            return ops.linear_bias(x, self.weight, self.bias)
        return torch.matmul(x, self.weight.t())

    def _use_fp8(self, x):
        return _fp8_ok(x, self.weight, self)


class RowParallelLinear(nn.Module):
    """Y = X @ W^T + b with W column-sharded over in_features; input arrives
    feature-sharded, output is all-reduced over the tp axis."""

    def __init__(self, in_features: int, out_features: int,
                 mesh: Optional[DeviceMesh] = None, axis: int = 1,
                 bias: bool = True, dtype=torch.float32, device=None,
                 init_seed=None, init_tag: str = None):
        super().__init__()
        self.mesh, self.axis = mesh, axis
        tp = mesh.axis_size(axis) if mesh is not None else 1
        assert in_features % tp == 0
        self.in_features = in_features
        self.out_features = out_features
        self.in_per_rank = in_features // tp
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_per_rank, dtype=dtype,
                        device=device))
        idx = mesh.axis_index(axis) if (mesh is not None and mesh.is_member) else 0
        seed = tag_seed(init_seed, init_tag) if init_tag is not None else None
        _sharded_normal_(self.weight, (out_features, in_features), 1,
                         max(idx, 0), tp, 1.0 / math.sqrt(in_features), seed)
        if bias:
            self.bias = nn.Parameter(
                torch.zeros(out_features, dtype=dtype, device=device))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        if ops.skinny_ok(x, self.weight, self):
            y = ops.skinny_linear(x, self.weight, None, self)
        elif _fp8_ok(x, self.weight, self):
            from ..ops.fp8 import fp8_linear
            y = fp8_linear(x, self.weight, module=self)
        elif (self.mesh is None or self.mesh.axis_size(self.axis) == 1) \
                and self.bias is not None:
            # tp degenerate: no all-reduce between GEMM and bias, so the
            # bias fuses into the hipBLASLt epilogue
            return ops.linear_bias(x, self.weight, self.bias)
        else:
            y = torch.matmul(x, self.weight.t())
        y = reduce_from_tp(y, self.mesh, self.axis)
        if self.bias is not None:
            y = ops.bias_add(y, self.bias)  # after the tp all-reduce
        return y


class VocabParallelEmbedding(nn.Module):
    """Embedding with the vocab dim sharded over the tp axis.

    Out-of-shard ids produce zeros; the all-reduce combines shards.
    """

    def __init__(self, num_embeddings: int, embedding_dim: int,
                 mesh: Optional[DeviceMesh] = None, axis: int = 1,
                 dtype=torch.float32, device=None, init_seed=None,
                 init_tag: str = None):
        super().__init__()
        self.mesh, self.axis = mesh, axis
        tp = mesh.axis_size(axis) if mesh is not None else 1
        assert num_embeddings % tp == 0
        self.num_embeddings = num_embeddings
        self.vocab_per_rank = num_embeddings // tp
        idx = mesh.axis_index(axis) if (mesh is not None and mesh.is_member) else 0
        idx = max(idx, 0)
        self.vocab_start = idx * self.vocab_per_rank
        self.weight = nn.Parameter(
            torch.empty(self.vocab_per_rank, embedding_dim, dtype=dtype,
                        device=device))
        seed = tag_seed(init_seed, init_tag) if init_tag is not None else None
        _sharded_normal_(self.weight, (num_embeddings, embedding_dim), 0,
                         idx, tp, 0.02, seed)

    def forward(self, ids: torch.Tensor):
        if self.mesh is None or self.mesh.axis_size(self.axis) == 1:
            return nn.functional.embedding(ids, self.weight)
        local = ids - self.vocab_start
        mask = (local < 0) | (local >= self.vocab_per_rank)
        local = local.clamp(0, self.vocab_per_rank - 1)
        y = nn.functional.embedding(local, self.weight)
        y = y.masked_fill(mask.unsqueeze(-1), 0.0)
        return reduce_from_tp(y, self.mesh, self.axis)


class _VocabParallelCrossEntropy(torch.autograd.Function):
    """Cross-entropy over vocab-sharded logits [N, V/tp] without gathering.

    Forward: global max + sum-exp via two all-reduces; target logit fetched
    from whichever rank owns it.  Backward: dlogits = (softmax - onehot)*dl,
    computed locally from saved (logits, global lse).
    """

    @staticmethod
    def forward(ctx, logits, targets, mesh: DeviceMesh, axis: int,
                vocab_start: int):
        vpr = logits.shape[-1]
        lf = logits.float()
        # global max
        mx = lf.max(dim=-1).values
        mesh.all_reduce(mx, axis=axis, op=dist.ReduceOp.MAX)
        # global sum of exp
        sumexp = torch.exp(lf - mx.unsqueeze(-1)).sum(dim=-1)
        mesh.all_reduce(sumexp, axis=axis)
        lse = mx + torch.log(sumexp)
        # target logit (owned by exactly one rank along the axis)
        local_t = targets - vocab_start
        in_shard = (local_t >= 0) & (local_t < vpr)
        t_idx = local_t.clamp(0, vpr - 1)
        t_logit = lf.gather(-1, t_idx.unsqueeze(-1)).squeeze(-1)
        t_logit = torch.where(in_shard, t_logit, torch.zeros_like(t_logit))
        mesh.all_reduce(t_logit, axis=axis)
        loss = lse - t_logit
        ctx.save_for_backward(logits, lse, t_idx, in_shard)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, lse, t_idx, in_shard = ctx.saved_tensors
        p = torch.exp(logits.float() - lse.unsqueeze(-1))
        upd = torch.where(in_shard, torch.ones_like(lse), torch.zeros_like(lse))
        p.scatter_add_(-1, t_idx.unsqueeze(-1), -upd.unsqueeze(-1))
        dlogits = (p * dloss.float().unsqueeze(-1)).to(logits.dtype)
        return dlogits, None, None, None, None


def vocab_parallel_cross_entropy(logits: torch.Tensor, targets: torch.Tensor,
                                 mesh: Optional[DeviceMesh], axis: int,
                                 vocab_start: int) -> torch.Tensor:
    """Per-token loss for vocab-sharded logits. Falls back to the fused
    single-device kernel when tp == 1."""
    if mesh is None or mesh.axis_size(axis) == 1:
        return ops.softmax_cross_entropy(logits, targets)
    return _VocabParallelCrossEntropy.apply(logits, targets, mesh, axis,
                                            vocab_start)


class _GatherFromParallel(torch.autograd.Function):
    """All-gather the feature (last) dim forward; slice backward (the
    Megatron gather/split conjugate pair — downstream computation is
    replicated SPMD, so each rank's grad of the full tensor is identical
    and backward just takes this rank's slice)."""

    @staticmethod
    def forward(ctx, x, mesh: DeviceMesh, axis: int):
        ctx.mesh, ctx.axis = mesh, axis
        tp = mesh.axis_size(axis)
        x = x.contiguous()
        parts = [torch.empty_like(x) for _ in range(tp)]
        dist.all_gather(parts, x, group=mesh.axis_group(axis))
        return torch.cat(parts, dim=-1)

    @staticmethod
    def backward(ctx, g):
        tp = ctx.mesh.axis_size(ctx.axis)
        idx = ctx.mesh.axis_index(ctx.axis)
        n = g.shape[-1] // tp
        return g.narrow(-1, idx * n, n).contiguous(), None, None


class _ScatterToParallel(torch.autograd.Function):
    """Slice this rank's feature shard forward; all-gather backward
    (Megatron scatter/gather conjugate: the producer is replicated, its
    grad is the concatenation of the per-shard consumer grads)."""

    @staticmethod
    def forward(ctx, x, mesh: DeviceMesh, axis: int):
        ctx.mesh, ctx.axis = mesh, axis
        tp = mesh.axis_size(axis)
        idx = mesh.axis_index(axis)
        n = x.shape[-1] // tp
        return x.narrow(-1, idx * n, n).contiguous()

    @staticmethod
    def backward(ctx, g):
        tp = ctx.mesh.axis_size(ctx.axis)
        g = g.contiguous()
        parts = [torch.empty_like(g) for _ in range(tp)]
        dist.all_gather(parts, g, group=ctx.mesh.axis_group(ctx.axis))
        return torch.cat(parts, dim=-1), None, None


def gather_from_tp(x, mesh, axis):
    """feature-sharded -> replicated (resharding edge (b,w) -> (b,None))."""
    if mesh is None or mesh.axis_size(axis) == 1:
        return x
    return _GatherFromParallel.apply(x, mesh, axis)


def scatter_to_tp(x, mesh, axis):
    """replicated -> feature-sharded (resharding edge (b,None) -> (b,w))."""
    if mesh is None or mesh.axis_size(axis) == 1:
        return x
    return _ScatterToParallel.apply(x, mesh, axis)


class _GatherFromParallelDim(torch.autograd.Function):
    """All-gather along an arbitrary dim forward; slice backward (the
    channel-dim variant of the Megatron gather/split pair — conv
    activations shard dim 1)."""

    @staticmethod
    def forward(ctx, x, mesh: DeviceMesh, axis: int, dim: int):
        ctx.mesh, ctx.axis, ctx.dim = mesh, axis, dim
        tp = mesh.axis_size(axis)
        x = x.contiguous()
        parts = [torch.empty_like(x) for _ in range(tp)]
        dist.all_gather(parts, x, group=mesh.axis_group(axis))
        return torch.cat(parts, dim=dim)

    @staticmethod
    def backward(ctx, g):
        tp = ctx.mesh.axis_size(ctx.axis)
        idx = ctx.mesh.axis_index(ctx.axis)
        n = g.shape[ctx.dim] // tp
        return (g.narrow(ctx.dim, idx * n, n).contiguous(), None, None,
                None)


class _ScatterToParallelDim(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, mesh: DeviceMesh, axis: int, dim: int):
        ctx.mesh, ctx.axis, ctx.dim = mesh, axis, dim
        tp = mesh.axis_size(axis)
        idx = mesh.axis_index(axis)
        n = x.shape[dim] // tp
        return x.narrow(dim, idx * n, n).contiguous()

    @staticmethod
    def backward(ctx, g):
        tp = ctx.mesh.axis_size(ctx.axis)
        g = g.contiguous()
        parts = [torch.empty_like(g) for _ in range(tp)]
        dist.all_gather(parts, g, group=ctx.mesh.axis_group(ctx.axis))
        return torch.cat(parts, dim=ctx.dim), None, None, None


def gather_from_tp_dim(x, mesh, axis, dim):
    if mesh is None or mesh.axis_size(axis) == 1:
        return x
    return _GatherFromParallelDim.apply(x, mesh, axis, dim)


def scatter_to_tp_dim(x, mesh, axis, dim):
    if mesh is None or mesh.axis_size(axis) == 1:
        return x
    return _ScatterToParallelDim.apply(x, mesh, axis, dim)


class ColumnParallelConv2d(nn.Module):
    """Conv2d with OUT channels sharded over the tp axis (the conv
    analog of column-parallel: dW strategies for conv, reference
    auto_sharding_dot_handler conv handling).  Output stays
    channel-sharded (dim 1); per-channel consumers (BN/ReLU/pool)
    operate on the shard unchanged."""

    def __init__(self, conv: nn.Conv2d, mesh: DeviceMesh, axis: int = 1):
        super().__init__()
        tp = mesh.axis_size(axis)
        idx = max(mesh.axis_index(axis), 0) if mesh.is_member else 0
        assert conv.out_channels % tp == 0 and conv.groups == 1
        self.mesh, self.axis = mesh, axis
        self.stride, self.padding = conv.stride, conv.padding
        self.dilation = conv.dilation
        per = conv.out_channels // tp
        with torch.no_grad():
            self.weight = nn.Parameter(
                conv.weight[idx * per:(idx + 1) * per].clone())
            self.bias = nn.Parameter(
                conv.bias[idx * per:(idx + 1) * per].clone()) \
                if conv.bias is not None else None
        self.out_channels, self.in_channels = conv.out_channels, \
            conv.in_channels

    def forward(self, x):
        x = copy_to_tp(x, self.mesh, self.axis)
        return nn.functional.conv2d(x, self.weight, self.bias,
                                    self.stride, self.padding,
                                    self.dilation)


class RowParallelConv2d(nn.Module):
    """Conv2d with IN channels sharded; input arrives channel-sharded,
    output partial sums all-reduce over the tp axis."""

    def __init__(self, conv: nn.Conv2d, mesh: DeviceMesh, axis: int = 1):
        super().__init__()
        tp = mesh.axis_size(axis)
        idx = max(mesh.axis_index(axis), 0) if mesh.is_member else 0
        assert conv.in_channels % tp == 0 and conv.groups == 1
        self.mesh, self.axis = mesh, axis
        self.stride, self.padding = conv.stride, conv.padding
        self.dilation = conv.dilation
        per = conv.in_channels // tp
        with torch.no_grad():
            self.weight = nn.Parameter(
                conv.weight[:, idx * per:(idx + 1) * per].clone())
            self.bias = nn.Parameter(conv.bias.clone()) \
                if conv.bias is not None else None
        self.out_channels, self.in_channels = conv.out_channels, \
            conv.in_channels

    def forward(self, x):
        y = nn.functional.conv2d(x, self.weight, None, self.stride,
                                 self.padding, self.dilation)
        y = reduce_from_tp(y, self.mesh, self.axis)
        if self.bias is not None:
            y = y + self.bias.view(1, -1, 1, 1)
        return y
