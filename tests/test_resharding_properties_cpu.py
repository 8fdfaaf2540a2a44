"""Property-based tests (hypothesis) for the cross-mesh resharding tile
algebra: ANY (src placement, dst placement) pair over the same global
shape must produce a transfer plan that exactly reconstructs every
destination tile (reference dst_tile_to_src_tiles_map,
cross_mesh_resharding.py:718 — the correctness core of pipeline-boundary
and restore-time resharding)."""
import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from alpa_amd.parallel.resharding import (Placement, ReshardingTaskSpec,
                                          _slice)


def _divisors(n):
    return [d for d in range(1, n + 1) if n % d == 0]


@st.composite
def placement_pair(draw):
    d0 = draw(st.sampled_from([4, 6, 8, 12]))
    d1 = draw(st.sampled_from([4, 6, 8]))
    shape = (d0, d1)

    def one(total_ranks):
        p0 = draw(st.sampled_from(_divisors(d0)))
        p1 = draw(st.sampled_from(_divisors(d1)))
        n_tiles = p0 * p1
        reps = draw(st.sampled_from([1, 2]))
        ranks = draw(st.permutations(range(n_tiles * reps)))
        return Placement(shape, (p0, p1), tuple(ranks))

    return shape, one(None), one(None)


def _simulate(spec: ReshardingTaskSpec, full: torch.Tensor):
    """Apply every transfer against per-rank buffers (single process
    simulation of the p2p exchange)."""
    src_bufs = {}
    for idx, owners in spec.src.tiles():
        t = _slice(full, idx).clone()
        for r in owners:
            src_bufs[r] = (idx, t)
    dst_bufs = {}
    for idx, owners in spec.dst.tiles():
        shape = tuple(hi - lo for lo, hi in idx)
        for r in owners:
            dst_bufs[r] = (idx, torch.full(shape, float("nan")))
    for t in spec.transfers:
        _, src_t = src_bufs[t.src_rank]
        _, dst_t = dst_bufs[t.dst_rank]
        _slice(dst_t, t.dst_offset).copy_(_slice(src_t, t.src_offset))
    return dst_bufs


@settings(max_examples=60, deadline=None)
@given(placement_pair())
def test_any_resharding_reconstructs_exactly(pair):
    shape, src, dst = pair
    torch.manual_seed(0)
    full = torch.randn(*shape)
    spec = ReshardingTaskSpec.build(src, dst)
    dst_bufs = _simulate(spec, full)
    for r, (idx, buf) in dst_bufs.items():
        assert not torch.isnan(buf).any(), (r, idx)
        torch.testing.assert_close(buf, _slice(full, idx))


@settings(max_examples=30, deadline=None)
@given(placement_pair())
def test_transfer_plan_is_minimal_per_destination(pair):
    """No destination element is written twice by DIFFERENT regions
    (each dst coordinate covered exactly once)."""
    shape, src, dst = pair
    spec = ReshardingTaskSpec.build(src, dst)
    per_dst = {}
    for t in spec.transfers:
        cover = per_dst.setdefault(t.dst_rank, np.zeros(
            tuple(hi - lo for lo, hi in spec.dst.rank_tile(t.dst_rank)),
            dtype=int))
        sl = tuple(slice(lo, hi) for lo, hi in t.dst_offset)
        cover[sl] += 1
    for r, cover in per_dst.items():
        assert cover.max() <= 1, r
        assert cover.min() >= 1, r
