"""Ulysses sequence-parallel attention tests: sp=2 over gloo vs serial
(the SURVEY §5.7 extension — absent in the reference)."""
import pytest
import torch

from dist_utils import run_distributed

import alpa_amd as aa
from alpa_amd.parallel.sequence import UlyssesAttention, shard_sequence

B, S, HID, HEADS = 2, 32, 64, 4


def build(mesh=None, sp_axis=1):
    return UlyssesAttention(HID, HEADS, mesh, sp_axis, init_seed=9)


def make_x(seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(B, S, HID, generator=g)


def test_serial_matches_plain_attention():
    m = build()
    x = make_x()
    y = m(x)
    assert y.shape == (B, S, HID)
    y.sum().backward()
    assert m.qkv.weight.grad is not None


def _sp_worker(rank, world_size):
    mesh = aa.full_mesh((1, world_size))
    m = build(mesh, 1)
    x = make_x()
    xl = shard_sequence(x, world_size, mesh.axis_index(1))
    xl = xl.clone().requires_grad_(True)
    y = m(xl)
    y.square().mean().backward()
    return y.detach(), xl.grad


def test_sp2_matches_serial():
    """Forward AND input-grad equality of the sequence-sharded attention
    against the serial module (same tag-seeded weights)."""
    m = build()
    x = make_x().requires_grad_(True)
    y = m(x)
    per = S // 2
    # every rank backprops its local per-shard mean loss; the all-to-all
    # backwards mix the contributions, so each rank's input grad equals
    # the serial grad of the SUM of the per-shard losses
    total = sum(y[:, r * per:(r + 1) * per].square().mean()
                for r in range(2))
    total.backward()
    results = run_distributed(_sp_worker, world_size=2, timeout=300)
    for r, (y_shard, xg) in enumerate(results):
        expect = y[:, r * per:(r + 1) * per]
        torch.testing.assert_close(y_shard, expect.detach(), rtol=1e-4,
                                   atol=1e-5)
        torch.testing.assert_close(
            xg, x.grad[:, r * per:(r + 1) * per], rtol=1e-4, atol=1e-5)


# ---------------------------------------------------------------------------
# Ring-attention module (context parallelism — the second SP mode)
# ---------------------------------------------------------------------------


def build_ring(mesh=None, sp_axis=1):
    from alpa_amd.parallel.sequence import RingSelfAttention
    return RingSelfAttention(HID, HEADS, mesh, sp_axis, init_seed=9)


def test_ring_module_serial_matches_ulysses_serial():
    """Both SP modes degenerate to the same plain attention at sp=1 (same
    tag-seeded weights)."""
    x = make_x()
    y_r = build_ring()(x)
    y_u = build()(x)
    torch.testing.assert_close(y_r, y_u, rtol=1e-5, atol=1e-5)


def _ring_sp_worker(rank, world_size):
    mesh = aa.full_mesh((1, world_size))
    m = build_ring(mesh, 1)
    x = make_x()
    xl = shard_sequence(x, world_size, mesh.axis_index(1))
    xl = xl.clone().requires_grad_(True)
    y = m(xl)
    y.square().mean().backward()
    return y.detach(), xl.grad, m.qkv.weight.grad


def test_ring_sp2_matches_serial():
    """Ring-attention SP: forward, input grads AND weight grads equal the
    serial module under the sum-of-per-shard losses."""
    m = build_ring()
    x = make_x().requires_grad_(True)
    y = m(x)
    per = S // 2
    total = sum(y[:, r * per:(r + 1) * per].square().mean()
                for r in range(2))
    total.backward()
    results = run_distributed(_ring_sp_worker, world_size=2, timeout=300)
    for r, (y_shard, xg, wg) in enumerate(results):
        expect = y[:, r * per:(r + 1) * per]
        torch.testing.assert_close(y_shard, expect.detach(), rtol=1e-4,
                                   atol=1e-5)
        torch.testing.assert_close(
            xg, x.grad[:, r * per:(r + 1) * per], rtol=1e-4, atol=1e-5)


def _zz_module_worker(rank, world_size):
    from alpa_amd.parallel.ring_attention import shard_zigzag
    from alpa_amd.parallel.sequence import RingSelfAttention
    mesh = aa.full_mesh((1, world_size))
    m = RingSelfAttention(HID, HEADS, mesh, 1, init_seed=9, zigzag=True)
    x = make_x()
    xl = shard_zigzag(x, mesh, axis=1, dim=1).requires_grad_(True)
    y = m(xl)
    y.square().mean().backward()
    return y.detach()


def test_zigzag_module_matches_serial():
    from alpa_amd.parallel.sequence import RingSelfAttention
    m = RingSelfAttention(HID, HEADS, None, 1, init_seed=9)
    x = make_x()
    y = m(x)
    results = run_distributed(_zz_module_worker, world_size=2, timeout=300)
    per = S // 4
    for r, ys in enumerate(results):
        lo = y[:, r * per:(r + 1) * per]
        hi = y[:, (3 - r) * per:(4 - r) * per]
        want = torch.cat([lo, hi], dim=1)
        torch.testing.assert_close(torch.as_tensor(ys), want.detach(),
                                   rtol=1e-4, atol=1e-5)
