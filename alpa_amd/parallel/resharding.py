"""Cross-mesh resharding: tile algebra + send/recv plan + execution.

Capability analog of the reference's ``pipeline_parallel/
cross_mesh_resharding.py`` (1903 LoC) and ``resharding_tensor.py``:
`VirtualDistributedArray:25` (sharding spec -> per-device tiles without
materialization), `ReshardingTaskSpec:674` with
`dst_tile_to_src_tiles_map:718` (tile intersection algebra), and the
send/recv strategy (`SymbolicReshardingTask:184` compiles per-worker
send/recv tile lists).

Used when a tensor sharded over one mesh (src placement) must move to a
different sharding on another (or the same) mesh — heterogeneous pipeline
stage boundaries, plan changes on restore, mesh resizing.  Execution is
RCCL/gloo p2p: every (src_rank, dst_rank) pair exchanges exactly the
overlap of their tiles; deterministic task order (sorted by rank pair)
keeps the exchange deadlock-free (reference sorts strategy order,
SURVEY.md §5.2).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch
import torch.distributed as dist

from ..mesh import is_distributed, rank

Index = Tuple[Tuple[int, int], ...]  # per-dim (start, stop)


@dataclass(frozen=True)
class Placement:
    """A tensor distributed over `ranks`: dim_partitions[d] = how many ways
    dim d is split; rank_grid maps partition coordinates -> global rank.

    This is the executable core of the reference's sharding-spec ->
    tile mapping (VirtualDistributedArray, resharding_tensor.py:25).
    """
    global_shape: Tuple[int, ...]
    dim_partitions: Tuple[int, ...]
    ranks: Tuple[int, ...]  # row-major over partition coords; replicas allowed

    def __post_init__(self):
        n_tiles = int(np.prod(self.dim_partitions))
        assert len(self.ranks) % n_tiles == 0, \
            (len(self.ranks), self.dim_partitions)

    @property
    def n_replicas(self) -> int:
        return len(self.ranks) // int(np.prod(self.dim_partitions))

    def tile_index(self, coord: Sequence[int]) -> Index:
        out = []
        for d, c in enumerate(coord):
            n = self.dim_partitions[d]
            size = self.global_shape[d]
            assert size % n == 0, (size, n)
            per = size // n
            out.append((c * per, (c + 1) * per))
        return tuple(out)

    def tiles(self) -> List[Tuple[Index, List[int]]]:
        """[(index, [owner ranks (replicas)])] in row-major coord order."""
        coords = np.indices(self.dim_partitions).reshape(
            len(self.dim_partitions), -1).T
        n_tiles = len(coords)
        out = []
        for i, c in enumerate(coords):
            owners = [self.ranks[r * n_tiles + i]
                      for r in range(self.n_replicas)]
            out.append((self.tile_index(c), owners))
        return out

    def rank_tile(self, r: int) -> Optional[Index]:
        for idx, owners in self.tiles():
            if r in owners:
                return idx
        return None


def _pick_replica(owners: Sequence[int], region: Index,
                  load: Dict[int, int]) -> int:
    """Greedy min-load source-replica choice: assign this transfer to
    the replica with the fewest bytes scheduled so far (ties broken by
    rank for determinism).  Balances per-link xGMI traffic the way the
    reference's randomized-greedy task solver does
    (cross_mesh_resharding.py:1615), but deterministically — every rank
    must compute the identical spec."""
    n = int(np.prod([hi - lo for lo, hi in region]))
    best = min(owners, key=lambda r: (load.get(r, 0), r))
    load[best] = load.get(best, 0) + n
    return best


def _intersect(a: Index, b: Index) -> Optional[Index]:
    out = []
    for (a0, a1), (b0, b1) in zip(a, b):
        lo, hi = max(a0, b0), min(a1, b1)
        if lo >= hi:
            return None
        out.append((lo, hi))
    return tuple(out)


@dataclass(frozen=True)
class TileTransfer:
    src_rank: int
    dst_rank: int
    region: Index        # global coordinates of the overlap
    src_offset: Index    # region relative to the src tile
    dst_offset: Index    # region relative to the dst tile


@dataclass(frozen=True)
class AllGatherFix:
    """Post-exchange intra-group all-gather for one replicated dst tile
    (the scatter-allgather rewrite, reference _rewrite_allgather_spec,
    cross_mesh_resharding.py:995): the p2p phase delivers each replica
    only 1/R of the tile; the replicas then all-gather locally."""
    owners: Tuple[int, ...]
    split_dim: int


def _replica_group_available(owners: Tuple[int, ...]) -> bool:
    """scatter-allgather needs a pre-created process group over the
    replica set; DeviceMesh construction creates every row/column group
    on every rank, so boundary replica sets are always cached.  Unknown
    sets fall back to full-replica sends (creating a group here would be
    a collective only a subset reaches)."""
    from ..mesh import _GROUP_CACHE, is_distributed as _isd
    if not _isd():
        return False
    return tuple(owners) in _GROUP_CACHE


@dataclass
class ReshardingTaskSpec:
    """All transfers needed to convert src placement -> dst placement
    (reference ReshardingTaskSpec + dst_tile_to_src_tiles_map)."""
    src: Placement
    dst: Placement
    transfers: List[TileTransfer]
    ag_fixes: List[AllGatherFix] = None

    @staticmethod
    def build(src: Placement, dst: Placement,
              scatter_allgather: bool = False) -> "ReshardingTaskSpec":
        """scatter_allgather: for dst tiles with R replicas, ship 1/R of
        the tile to each replica and all-gather inside the replica group
        afterwards — cross-placement traffic drops by R (reference local
        all-gather optimization, use_local_allgather)."""
        assert src.global_shape == dst.global_shape
        transfers = []
        ag_fixes = []
        load: Dict[int, int] = {}
        src_tiles = src.tiles()
        for dst_idx, dst_owners in dst.tiles():
            R = len(dst_owners)
            sa_dim = None
            if scatter_allgather and R > 1 and \
                    _replica_group_available(tuple(dst_owners)):
                dims = sorted(range(len(dst_idx)),
                              key=lambda d: -(dst_idx[d][1] - dst_idx[d][0]))
                for d in dims:
                    if (dst_idx[d][1] - dst_idx[d][0]) % R == 0 and \
                            (dst_idx[d][1] - dst_idx[d][0]) >= R:
                        sa_dim = d
                        break
            if sa_dim is not None:
                lo0, hi0 = dst_idx[sa_dim]
                per = (hi0 - lo0) // R
                for r_i, d_own in enumerate(dst_owners):
                    sub = list(dst_idx)
                    sub[sa_dim] = (lo0 + r_i * per, lo0 + (r_i + 1) * per)
                    sub = tuple(sub)
                    for src_idx, src_owners in src_tiles:
                        inter = _intersect(sub, src_idx)
                        if inter is None:
                            continue
                        s_own = _pick_replica(src_owners, inter, load)
                        transfers.append(TileTransfer(
                            src_rank=s_own, dst_rank=d_own, region=inter,
                            src_offset=tuple(
                                (lo - s0, hi - s0) for (lo, hi),
                                (s0, _) in zip(inter, src_idx)),
                            dst_offset=tuple(
                                (lo - d0, hi - d0) for (lo, hi),
                                (d0, _) in zip(inter, dst_idx))))
                ag_fixes.append(AllGatherFix(tuple(dst_owners), sa_dim))
                continue
            for d_own in dst_owners:
                for src_idx, src_owners in src_tiles:
                    inter = _intersect(dst_idx, src_idx)
                    if inter is None:
                        continue
                    # greedy load-balanced replica choice: the replica
                    # with the least bytes assigned so far sends
                    # (deterministic; reference's randomized-greedy /
                    # DFS task solvers, cross_mesh_resharding.py:1615)
                    s_own = _pick_replica(src_owners, inter, load)
                    transfers.append(TileTransfer(
                        src_rank=s_own, dst_rank=d_own, region=inter,
                        src_offset=tuple((lo - s0, hi - s0) for (lo, hi),
                                         (s0, _) in zip(inter, src_idx)),
                        dst_offset=tuple((lo - d0, hi - d0) for (lo, hi),
                                         (d0, _) in zip(inter, dst_idx))))
        # deterministic global order => deadlock-free paired exchange
        transfers.sort(key=lambda t: (t.src_rank, t.dst_rank, t.region))
        return ReshardingTaskSpec(src, dst, transfers, ag_fixes)

    def total_bytes(self, elem_size: int = 2) -> int:
        n = 0
        for t in self.transfers:
            if t.src_rank != t.dst_rank:
                n += int(np.prod([hi - lo for lo, hi in t.region])) * elem_size
        return n


def _slice(t: torch.Tensor, idx: Index) -> torch.Tensor:
    return t[tuple(slice(lo, hi) for lo, hi in idx)]


def prepare_resharding(spec: ReshardingTaskSpec,
                       local_src: Optional[torch.Tensor],
                       dst_buf: Optional[torch.Tensor]):
    """This rank's part of the exchange, split for composability: performs
    the local copies now and returns (p2p_ops, fixups); the caller batches
    the ops (possibly merged with another exchange) in one
    batch_isend_irecv and then applies the fixups."""
    me = rank()
    for t in spec.transfers:
        if t.src_rank == me and t.dst_rank == me:
            _slice(dst_buf, t.dst_offset).copy_(_slice(local_src,
                                                       t.src_offset))
    ops: List[dist.P2POp] = []
    fixups = []
    if not is_distributed():
        return ops, fixups
    for t in spec.transfers:
        if t.src_rank == t.dst_rank:
            continue
        if t.src_rank == me:
            payload = _slice(local_src, t.src_offset).contiguous()
            fixups.append((payload, None))  # keep alive until waited
            ops.append(dist.P2POp(dist.isend, payload, t.dst_rank))
        elif t.dst_rank == me:
            shape = tuple(hi - lo for lo, hi in t.region)
            buf = torch.empty(shape, dtype=dst_buf.dtype,
                              device=dst_buf.device)
            fixups.append((buf, (dst_buf, t)))
            ops.append(dist.P2POp(dist.irecv, buf, t.src_rank))
    return ops, fixups


def apply_fixups(fixups):
    for buf, tgt in fixups:
        if tgt is not None:
            dst_buf, t = tgt
            _slice(dst_buf, t.dst_offset).copy_(buf)


def apply_allgather_fixes(spec: ReshardingTaskSpec,
                          dst_buf: Optional[torch.Tensor]):
    """Scatter-allgather phase 2: replicas of each dst tile exchange
    their 1/R stripes (runs after the p2p fixups)."""
    if not spec.ag_fixes or dst_buf is None or not is_distributed():
        return
    from ..mesh import _GROUP_CACHE
    me = rank()
    for f in spec.ag_fixes:
        if me not in f.owners:
            continue
        g = _GROUP_CACHE.get(f.owners)
        R = len(f.owners)
        r = f.owners.index(me)
        n = dst_buf.shape[f.split_dim] // R
        mine = dst_buf.narrow(f.split_dim, r * n, n).contiguous()
        parts = [torch.empty_like(mine) for _ in range(R)]
        dist.all_gather(parts, mine, group=g)
        for j in range(R):
            if j != r:
                dst_buf.narrow(f.split_dim, j * n, n).copy_(parts[j])


def execute_resharding(spec: ReshardingTaskSpec,
                       local_src: Optional[torch.Tensor],
                       dst_buf: Optional[torch.Tensor]
                       ) -> Optional[torch.Tensor]:
    """Run this rank's part of the exchange (see prepare_resharding)."""
    ops, fixups = prepare_resharding(spec, local_src, dst_buf)
    if ops:
        for w in dist.batch_isend_irecv(ops):
            w.wait()
    apply_fixups(fixups)
    apply_allgather_fixes(spec, dst_buf)
    return dst_buf
