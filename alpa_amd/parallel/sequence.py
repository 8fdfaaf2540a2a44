"""Sequence parallelism: Ulysses-style all-to-all attention.

The reference has NO long-context support (no ring
attention, no Ulysses anywhere in alpa/; max benchmark seq 1024) — this is
the MI355X-native *extension* the survey plans: the residual stream stays
sequence-sharded [B, S/sp, H] across the sp group (activation memory and
every per-token op scale 1/sp); around attention, one all-to-all swaps the
sharding to head-sharded full-sequence [B, S, h/sp, d] (attention needs
the whole sequence per head), and a second all-to-all swaps back.  On the
xGMI full crossbar all-to-all is per-link parallel — the natural
collective for this exchange.
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..mesh import DeviceMesh
from .expert import all_to_all
from .layers import ColumnParallelLinear, RowParallelLinear


def _seq_to_head_shard(qkv: torch.Tensor, sp: int, mesh, axis,
                       heads: int, d3: int) -> torch.Tensor:
    """[B, S/sp, heads, d3] -> [B, S, heads/sp, d3] via all-to-all."""
    B, Sl, h, _ = qkv.shape
    hps = h // sp
    # order by destination rank: block r = our seq chunk of head group r
    blocks = qkv.view(B, Sl, sp, hps, d3).permute(2, 0, 1, 3, 4).contiguous()
    out = all_to_all(blocks, mesh, axis)  # [sp, B, S/sp, hps, d3]
    # concat received seq chunks: dim0 = source rank = seq block index
    return out.permute(1, 0, 2, 3, 4).reshape(B, sp * Sl, hps, d3)


def _head_to_seq_shard(o: torch.Tensor, sp: int, mesh, axis,
                       heads: int, d: int) -> torch.Tensor:
    """[B, S, heads/sp, d] -> [B, S/sp, heads, d] via all-to-all."""
    B, S, hps, _ = o.shape
    Sl = S // sp
    blocks = o.view(B, sp, Sl, hps, d).permute(1, 0, 2, 3, 4).contiguous()
    out = all_to_all(blocks, mesh, axis)  # [sp, B, Sl, hps, d]
    return out.permute(1, 2, 0, 3, 4).reshape(B, Sl, sp * hps, d)


class UlyssesAttention(nn.Module):
    """Multi-head attention over a sequence-sharded residual stream.

    Input/output [B, S/sp, hidden]; internally two all-to-alls reshard
    seq<->head around the flash kernel.  TP (head sharding at rest) and SP
    can share or use different mesh axes; here sp rides one axis and the
    projections stay unsharded (sp == "the" model axis for this module).
    """

    def __init__(self, hidden: int, num_heads: int,
                 mesh: Optional[DeviceMesh] = None, sp_axis: int = 1,
                 dtype=torch.float32, device=None, layer_idx: int = 0,
                 init_seed: int = 0):
        super().__init__()
        self.mesh, self.sp_axis = mesh, sp_axis
        self.sp = mesh.axis_size(sp_axis) if mesh is not None else 1
        assert num_heads % max(self.sp, 1) == 0
        self.heads = num_heads
        self.head_dim = hidden // num_heads
        # projections replicated (grads sync over the dp axis as usual);
        # the sp sharding lives in the activations
        self.qkv = ColumnParallelLinear(hidden, 3 * hidden, None, 1,
                                        dtype=dtype, device=device,
                                        init_seed=init_seed,
                                        init_tag=f"sp{layer_idx}.qkv")
        self.out = RowParallelLinear(hidden, hidden, None, 1, dtype=dtype,
                                     device=device, init_seed=init_seed,
                                     init_tag=f"sp{layer_idx}.out")

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x [B, S/sp, hidden] (the rank's seq shard, in order of rank)."""
        B, Sl, _ = x.shape
        h, d = self.heads, self.head_dim
        qkv = self.qkv(x).view(B, Sl, h, 3 * d)
        if self.sp > 1:
            qkv = _seq_to_head_shard(qkv, self.sp, self.mesh, self.sp_axis,
                                     h, 3 * d)  # [B, S, h/sp, 3d]
        o = ops.flash_attention_qkv(
            qkv.reshape(B, qkv.shape[1], -1), qkv.shape[2], causal=True)
        if self.sp > 1:
            o = o.view(B, o.shape[1], h // self.sp, d)
            o = _head_to_seq_shard(o, self.sp, self.mesh, self.sp_axis, h, d)
            o = o.reshape(B, Sl, h * d)
        return self.out(o)


class RingSelfAttention(nn.Module):
    """Multi-head attention over a sequence-sharded residual stream using
    RING attention (parallel/ring_attention.py) instead of Ulysses'
    all-to-alls: heads stay whole, K/V chunks stream around the ring and
    partials merge via lse — per-rank peak memory stays O(S/sp) even for
    the attention working set, the memory-optimal long-context mode.
    Drop-in interface match with UlyssesAttention ([B, S/sp, hidden])."""

    def __init__(self, hidden: int, num_heads: int,
                 mesh: Optional[DeviceMesh] = None, sp_axis: int = 1,
                 dtype=torch.float32, device=None, layer_idx: int = 0,
                 init_seed: int = 0, zigzag: bool = False):
        super().__init__()
        self.mesh, self.sp_axis = mesh, sp_axis
        self.sp = mesh.axis_size(sp_axis) if mesh is not None else 1
        self.heads = num_heads
        self.head_dim = hidden // num_heads
        #: zigzag mode: the residual stream is shard_zigzag-sharded
        #: (rank i holds chunks i and 2n-1-i) — equal causal work per rank
        self.zigzag = zigzag
        self.qkv = ColumnParallelLinear(hidden, 3 * hidden, None, 1,
                                        dtype=dtype, device=device,
                                        init_seed=init_seed,
                                        init_tag=f"sp{layer_idx}.qkv")
        self.out = RowParallelLinear(hidden, hidden, None, 1, dtype=dtype,
                                     device=device, init_seed=init_seed,
                                     init_tag=f"sp{layer_idx}.out")

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x [B, S/sp, hidden] (this rank's seq chunk; rank-ordered, or
        shard_zigzag layout in zigzag mode)."""
        from .ring_attention import ring_attention, zigzag_ring_attention
        B, Sl, _ = x.shape
        h, d = self.heads, self.head_dim
        qkv = self.qkv(x).view(B, Sl, h, 3, d)
        q = qkv[:, :, :, 0].permute(0, 2, 1, 3).contiguous()
        k = qkv[:, :, :, 1].permute(0, 2, 1, 3).contiguous()
        v = qkv[:, :, :, 2].permute(0, 2, 1, 3).contiguous()
        if self.zigzag and self.sp > 1:
            o = zigzag_ring_attention(q, k, v, self.mesh, self.sp_axis)
        else:
            o = ring_attention(q, k, v, self.mesh, self.sp_axis,
                               causal=True)
        o = o.permute(0, 2, 1, 3).reshape(B, Sl, h * d)
        return self.out(o)


def shard_sequence(x: torch.Tensor, sp: int, idx: int,
                   dim: int = 1) -> torch.Tensor:
    """Slice a full-sequence tensor to this rank's seq shard."""
    S = x.shape[dim]
    assert S % sp == 0
    per = S // sp
    return x.narrow(dim, idx * per, per)
