"""Ring attention (context parallelism) vs serial flash attention.

EXTENSION beyond the reference (ring attention absent in
alpa); the serial oracle is the framework's own attention op, itself
tested against fp32 softmax attention in test_ops_cpu / GPU numerics.
"""
import pytest
import torch

from dist_utils import run_distributed

import alpa_amd as aa
from alpa_amd import ops
from alpa_amd.parallel.ring_attention import ring_attention, shard_ring

B, H, S, D = 2, 3, 32, 16


def _inputs(requires_grad=True):
    g = torch.Generator().manual_seed(42)
    q = torch.randn(B, H, S, D, generator=g)
    k = torch.randn(B, H, S, D, generator=g)
    v = torch.randn(B, H, S, D, generator=g)
    w = torch.cos(torch.arange(B * H * S * D, dtype=torch.float32)
                  ).reshape(B, H, S, D)
    for t in (q, k, v):
        t.requires_grad_(requires_grad)
    return q, k, v, w


def _serial(causal):
    q, k, v, w = _inputs()
    o = ops.flash_attention(q, k, v, causal=causal)
    (o * w).sum().backward()
    return o, q.grad, k.grad, v.grad


def _ring_worker(rank, world_size, causal):
    mesh = aa.mesh.full_mesh((1, world_size))
    q, k, v, w = _inputs(requires_grad=False)
    ql = shard_ring(q, mesh).requires_grad_(True)
    kl = shard_ring(k, mesh).requires_grad_(True)
    vl = shard_ring(v, mesh).requires_grad_(True)
    o = ring_attention(ql, kl, vl, mesh, axis=1, causal=causal)
    (o * shard_ring(w, mesh)).sum().backward()
    return (o.detach(), ql.grad, kl.grad, vl.grad)


@pytest.mark.parametrize("world_size,causal", [(2, True), (2, False),
                                               (4, True)])
def test_ring_matches_serial(world_size, causal):
    o_s, dq_s, dk_s, dv_s = _serial(causal)
    results = run_distributed(_ring_worker, world_size=world_size,
                              args=(causal,), timeout=300)
    per = S // world_size
    for r, (o, dq, dk, dv) in enumerate(results):
        sl = slice(r * per, (r + 1) * per)
        for got, want, name in (
                (o, o_s[:, :, sl], "o"), (dq, dq_s[:, :, sl], "dq"),
                (dk, dk_s[:, :, sl], "dk"), (dv, dv_s[:, :, sl], "dv")):
            torch.testing.assert_close(
                torch.as_tensor(got), want.detach(), rtol=2e-5, atol=2e-5,
                msg=lambda m: f"rank{r} {name}: {m}")


def test_ring_single_rank_fallback():
    q, k, v, w = _inputs()
    o = ring_attention(q, k, v, None, causal=True)
    (o * w).sum().backward()
    o_s, dq_s, dk_s, dv_s = _serial(True)
    torch.testing.assert_close(o, o_s, rtol=1e-6, atol=1e-6)
    torch.testing.assert_close(q.grad, dq_s, rtol=1e-6, atol=1e-6)


# ---------------------------------------------------------------------------
# Zigzag ring attention (causal load balance: rank i holds chunks i and
# 2n-1-i, so per-rank causal work is equal)
# ---------------------------------------------------------------------------


def _zz_positions(n, i, per):
    lo = slice(i * per, (i + 1) * per)
    hi = slice((2 * n - 1 - i) * per, (2 * n - i) * per)
    return lo, hi


def _zigzag_worker(rank, world_size):
    from alpa_amd.parallel.ring_attention import (shard_zigzag,
                                                  zigzag_ring_attention)
    mesh = aa.mesh.full_mesh((1, world_size))
    q, k, v, w = _inputs(requires_grad=False)
    ql = shard_zigzag(q, mesh).requires_grad_(True)
    kl = shard_zigzag(k, mesh).requires_grad_(True)
    vl = shard_zigzag(v, mesh).requires_grad_(True)
    o = zigzag_ring_attention(ql, kl, vl, mesh, axis=1)
    (o * shard_zigzag(w, mesh)).sum().backward()
    return (o.detach(), ql.grad, kl.grad, vl.grad)


@pytest.mark.parametrize("world_size", [2, 4])
def test_zigzag_matches_serial(world_size):
    o_s, dq_s, dk_s, dv_s = _serial(True)
    results = run_distributed(_zigzag_worker, world_size=world_size,
                              timeout=300)
    per = S // (2 * world_size)
    for r, (o, dq, dk, dv) in enumerate(results):
        lo, hi = _zz_positions(world_size, r, per)
        for got, want, name in (
                (o, o_s, "o"), (dq, dq_s, "dq"), (dk, dk_s, "dk"),
                (dv, dv_s, "dv")):
            want_cat = torch.cat([want[:, :, lo], want[:, :, hi]], dim=2)
            torch.testing.assert_close(
                torch.as_tensor(got), want_cat.detach(), rtol=2e-5,
                atol=2e-5, msg=lambda m: f"rank{r} {name}: {m}")
