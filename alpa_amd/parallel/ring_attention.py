"""Ring attention: context parallelism over the sequence axis.

EXTENSION beyond the reference (SURVEY.md §2.2 SP row: ring attention is
ABSENT in alpa — max seq len in its benchmarks is 1024; the survey
requires the new framework to design long-context support natively).

Each rank of the mesh axis holds a contiguous SEQUENCE chunk of Q, K, V
(`[B, heads, S/n, d]`).  K/V chunks travel around the ring (xGMI is a
full point-to-point crossbar — neighbor exchange uses one dedicated link
per direction while attention math runs); every rank computes partial
attention of its local queries against each arriving chunk with the
hand-written gfx950 flash kernel and merges partials with the standard
log-sum-exp rescaling — the kernel already emits per-row lse
(ops/csrc/attention.hip: lse_out) precisely so partials can be combined.

Backward re-rings the K/V chunks and accumulates (dK, dV) in fp32
accumulators that travel WITH their chunk; dQ stays local.  The per-chunk
backward reuses the fused attention backward (global o / lse / delta make
the chunk decomposition exact).

Causality across chunks: query global position = r·Sc + i, key position
= src·Sc + j ⇒ earlier chunks (src < r) attend fully, the own chunk is
standard causal, later chunks contribute nothing (ranks still forward
the ring so the exchange stays collective).
"""
from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from ..mesh import DeviceMesh, is_distributed
from ..ops import reference as ref
from ..ops._backend import hip_ops, use_hip


def _chunk_fwd(q, k, v, causal, scale):
    if use_hip(q):
        return hip_ops().attn_fwd(q.contiguous(), k.contiguous(),
                                  v.contiguous(), causal, scale)
    return ref.attention_fwd(q, k, v, causal, scale)


def _chunk_bwd(do, q, k, v, o, lse, causal, scale):
    if use_hip(q):
        return hip_ops().attn_bwd(do.contiguous(), q.contiguous(),
                                  k.contiguous(), v.contiguous(),
                                  o.contiguous(), lse.contiguous(),
                                  causal, scale)
    return ref.attention_bwd(do, q, k, v, o, lse, causal, scale)


def _ring_pass(peers: Tuple[int, ...], idx: int,
               tensors: List[torch.Tensor]) -> List[torch.Tensor]:
    """Send every tensor to the NEXT rank of `peers`, receive the same
    shapes from the PREV — one fused batch_isend_irecv (deadlock-free
    paired exchange, same discipline as the 1F1B engine's p2p)."""
    n = len(peers)
    nxt = peers[(idx + 1) % n]
    prv = peers[(idx - 1) % n]
    recvs = [torch.empty_like(t) for t in tensors]
    ops = []
    for t in tensors:
        ops.append(dist.P2POp(dist.isend, t.contiguous(), nxt))
    for b in recvs:
        ops.append(dist.P2POp(dist.irecv, b, prv))
    for w in dist.batch_isend_irecv(ops):
        w.wait()
    return recvs


def _merge(o_acc, lse_acc, o_t, lse_t):
    """Combine two normalized partials: o = Σ exp(lse_i - lse)·o_i.
    Rows with no attendable keys anywhere (both lse = -inf — cannot
    happen under causal self-attention, where every row sees at least
    itself) resolve to o = 0 / lse = -inf like the kernel's empty-row
    epilogue, instead of 0/0 NaNs."""
    m = torch.maximum(lse_acc, lse_t)
    dead = torch.isinf(m) & (m < 0)
    m_safe = torch.where(dead, torch.zeros_like(m), m)
    w1 = torch.exp(lse_acc - m_safe)
    w2 = torch.exp(lse_t - m_safe)
    l = (w1 + w2).clamp_min(1e-38)
    o = (w1.unsqueeze(-1) * o_acc + w2.unsqueeze(-1) * o_t.float()) / \
        l.unsqueeze(-1)
    o = torch.where(dead.unsqueeze(-1), torch.zeros_like(o), o)
    lse = torch.where(dead, torch.full_like(m, float("-inf")),
                      m_safe + torch.log(l))
    return o, lse


class _RingAttention(torch.autograd.Function):

    @staticmethod
    def forward(ctx, q, k, v, mesh: DeviceMesh, axis: int, causal, scale):
        if scale is None:
            scale = 1.0 / math.sqrt(q.shape[-1])
        n = mesh.axis_size(axis) if mesh is not None else 1
        if n == 1 or not is_distributed():
            o, lse = _chunk_fwd(q, k, v, causal, scale)
            ctx.save_for_backward(q, k, v, o, lse)
            ctx.meta = (mesh, axis, causal, scale, 1, 0, ())
            return o
        peers = mesh.axis_ranks(axis)
        r = mesh.axis_index(axis)
        k_cur, v_cur = k.contiguous(), v.contiguous()
        o_acc = torch.zeros(q.shape, dtype=torch.float32, device=q.device)
        lse_acc = torch.full(q.shape[:3], float("-inf"),
                             dtype=torch.float32, device=q.device)
        for t in range(n):
            src = (r - t) % n
            if not causal or src <= r:
                o_t, lse_t = _chunk_fwd(q, k_cur, v_cur,
                                        causal and src == r, scale)
                o_acc, lse_acc = _merge(o_acc, lse_acc, o_t,
                                        lse_t.float())
            if t < n - 1:
                k_cur, v_cur = _ring_pass(peers, r, [k_cur, v_cur])
        o = o_acc.to(q.dtype)
        ctx.save_for_backward(q, k, v, o, lse_acc)
        ctx.meta = (mesh, axis, causal, scale, n, r, peers)
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        mesh, axis, causal, scale, n, r, peers = ctx.meta
        if n == 1:
            dq, dk, dv = _chunk_bwd(do, q, k, v, o, lse, causal, scale)
            return dq, dk, dv, None, None, None, None
        k_cur, v_cur = k.contiguous(), v.contiguous()
        dq_acc = torch.zeros(q.shape, dtype=torch.float32, device=q.device)
        # (dk, dv) accumulators travel the ring WITH their chunk
        dk_acc = torch.zeros(k.shape, dtype=torch.float32, device=k.device)
        dv_acc = torch.zeros(v.shape, dtype=torch.float32, device=v.device)
        for t in range(n):
            src = (r - t) % n
            if not causal or src <= r:
                dq_t, dk_t, dv_t = _chunk_bwd(do, q, k_cur, v_cur, o, lse,
                                              causal and src == r, scale)
                dq_acc += dq_t.float()
                dk_acc += dk_t.float()
                dv_acc += dv_t.float()
            if t < n - 1:
                k_cur, v_cur, dk_acc, dv_acc = _ring_pass(
                    peers, r, [k_cur, v_cur, dk_acc, dv_acc])
        # the accumulators now hold grads for chunk (r+1)%n: one more hop
        # brings every chunk's grads home
        dk_acc, dv_acc = _ring_pass(peers, r, [dk_acc, dv_acc])
        return (dq_acc.to(q.dtype), dk_acc.to(k.dtype), dv_acc.to(v.dtype),
                None, None, None, None)


def ring_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   mesh: Optional[DeviceMesh], axis: int = 1,
                   causal: bool = True,
                   scale: Optional[float] = None) -> torch.Tensor:
    """Attention over a sequence sharded along `mesh` axis `axis`.
    q,k,v: this rank's chunk [B, heads, S/n, d]; returns the local output
    chunk.  Chunks must be contiguous slices in rank order."""
    return _RingAttention.apply(q, k, v, mesh, axis, causal, scale)


def shard_ring(x: torch.Tensor, mesh: Optional[DeviceMesh],
               axis: int = 1, dim: int = 2) -> torch.Tensor:
    """Slice this rank's contiguous sequence chunk (test/demo helper)."""
    if mesh is None or mesh.axis_size(axis) == 1:
        return x
    n = mesh.axis_size(axis)
    i = mesh.axis_index(axis)
    per = x.shape[dim] // n
    return x.narrow(dim, i * per, per).contiguous()


# ---------------------------------------------------------------------------
# Zigzag ring attention: causal load balancing
# ---------------------------------------------------------------------------


def shard_zigzag(x: torch.Tensor, mesh: Optional[DeviceMesh],
                 axis: int = 1, dim: int = 2) -> torch.Tensor:
    """Zigzag sharding: the sequence splits into 2n chunks; rank i holds
    the CONCAT of chunks i and 2n-1-i.  Under causal attention the two
    chunks' workloads sum to the same total on every rank — the plain
    ring's idle-rank-0 imbalance disappears."""
    if mesh is None or mesh.axis_size(axis) == 1:
        return x
    n = mesh.axis_size(axis)
    i = mesh.axis_index(axis)
    per = x.shape[dim] // (2 * n)
    a = x.narrow(dim, i * per, per)
    b = x.narrow(dim, (2 * n - 1 - i) * per, per)
    return torch.cat([a, b], dim=dim).contiguous()


def unshard_zigzag_grad_positions(n: int, i: int):
    """Chunk indices held by rank i under zigzag."""
    return (i, 2 * n - 1 - i)


def _zz_pair_fwd(q, kv_k, kv_v, my_c, in_c, scale):
    """Partial attention of one local q chunk against one arriving kv
    chunk with chunk-level causality (in_c < my_c: full; ==: causal;
    >: None)."""
    if in_c > my_c:
        return None
    return _chunk_fwd(q, kv_k, kv_v, in_c == my_c, scale)


class _ZigzagRingAttention(torch.autograd.Function):

    @staticmethod
    def forward(ctx, q, k, v, mesh: DeviceMesh, axis: int, scale):
        if scale is None:
            scale = 1.0 / math.sqrt(q.shape[-1])
        n = mesh.axis_size(axis) if mesh is not None else 1
        if n == 1 or not is_distributed():
            o, lse = _chunk_fwd(q, k, v, True, scale)
            ctx.save_for_backward(q, k, v, o, lse)
            ctx.meta = (mesh, axis, scale, 1, 0, ())
            return o
        peers = mesh.axis_ranks(axis)
        r = mesh.axis_index(axis)
        B, H, S2, D = q.shape
        Sc = S2 // 2
        my_cs = unshard_zigzag_grad_positions(n, r)
        qs = [q[:, :, :Sc], q[:, :, Sc:]]
        k_cur, v_cur = k.contiguous(), v.contiguous()
        o_acc = [torch.zeros(B, H, Sc, D, dtype=torch.float32,
                             device=q.device) for _ in range(2)]
        lse_acc = [torch.full((B, H, Sc), float("-inf"),
                              dtype=torch.float32, device=q.device)
                   for _ in range(2)]
        for t in range(n):
            src = (r - t) % n
            in_cs = unshard_zigzag_grad_positions(n, src)
            for qi in range(2):
                for ki in range(2):
                    res = _zz_pair_fwd(
                        qs[qi], k_cur[:, :, ki * Sc:(ki + 1) * Sc],
                        v_cur[:, :, ki * Sc:(ki + 1) * Sc],
                        my_cs[qi], in_cs[ki], scale)
                    if res is not None:
                        o_acc[qi], lse_acc[qi] = _merge(
                            o_acc[qi], lse_acc[qi], res[0],
                            res[1].float())
            if t < n - 1:
                k_cur, v_cur = _ring_pass(peers, r, [k_cur, v_cur])
        o = torch.cat(o_acc, dim=2).to(q.dtype)
        lse = torch.cat(lse_acc, dim=2)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.meta = (mesh, axis, scale, n, r, peers)
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        mesh, axis, scale, n, r, peers = ctx.meta
        if n == 1:
            dq, dk, dv = _chunk_bwd(do, q, k, v, o, lse, True, scale)
            return dq, dk, dv, None, None, None
        B, H, S2, D = q.shape
        Sc = S2 // 2
        my_cs = unshard_zigzag_grad_positions(n, r)
        k_cur, v_cur = k.contiguous(), v.contiguous()
        dq_acc = torch.zeros(q.shape, dtype=torch.float32, device=q.device)
        dk_acc = torch.zeros(k.shape, dtype=torch.float32, device=k.device)
        dv_acc = torch.zeros(v.shape, dtype=torch.float32, device=v.device)
        for t in range(n):
            src = (r - t) % n
            in_cs = unshard_zigzag_grad_positions(n, src)
            for qi in range(2):
                qsl = slice(qi * Sc, (qi + 1) * Sc)
                for ki in range(2):
                    if in_cs[ki] > my_cs[qi]:
                        continue
                    ksl = slice(ki * Sc, (ki + 1) * Sc)
                    dq_t, dk_t, dv_t = _chunk_bwd(
                        do[:, :, qsl], q[:, :, qsl], k_cur[:, :, ksl],
                        v_cur[:, :, ksl], o[:, :, qsl], lse[:, :, qsl],
                        in_cs[ki] == my_cs[qi], scale)
                    dq_acc[:, :, qsl] += dq_t.float()
                    dk_acc[:, :, ksl] += dk_t.float()
                    dv_acc[:, :, ksl] += dv_t.float()
            if t < n - 1:
                k_cur, v_cur, dk_acc, dv_acc = _ring_pass(
                    peers, r, [k_cur, v_cur, dk_acc, dv_acc])
        dk_acc, dv_acc = _ring_pass(peers, r, [dk_acc, dv_acc])
        return (dq_acc.to(q.dtype), dk_acc.to(k.dtype),
                dv_acc.to(v.dtype), None, None, None)


def zigzag_ring_attention(q: torch.Tensor, k: torch.Tensor,
                          v: torch.Tensor, mesh: Optional[DeviceMesh],
                          axis: int = 1,
                          scale: Optional[float] = None) -> torch.Tensor:
    """Causal ring attention with zigzag load balancing: q/k/v are this
    rank's shard_zigzag output ([B, h, 2*(S/2n), d]).  Always causal —
    that is what the balancing is for."""
    return _ZigzagRingAttention.apply(q, k, v, mesh, axis, scale)
