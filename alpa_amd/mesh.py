"""Device mesh runtime: one process per GPU over torch.distributed (RCCL/xGMI).

MI355X-native replacement for the reference's Ray-actor mesh runtime
(``alpa/device_mesh.py:107,979,1792,2131``).  There is no driver process and no
RPC: every GPU is a rank in a single ``torch.distributed`` world (backend
"nccl" == RCCL on ROCm, "gloo" for CPU-only tests), and all ranks execute the
same statically-compiled step program.  Logical meshes are views (shape +
axis process groups) over the flat rank world.

Key classes
-----------
- :class:`VirtualMesh` — an *unallocated* mesh used by the planners
  (analog of ``VirtualPhysicalMesh``, device_mesh.py:1792): shape + slicing,
  no process groups.
- :class:`DeviceMesh` — a live mesh: per-axis process groups, collectives,
  a dedicated comm stream for overlap (analog of
  ``DistributedPhysicalDeviceMesh``, device_mesh.py:979).
- :func:`init_distributed` / :func:`shutdown` — world bring-up/teardown
  (analog of ``alpa.init/shutdown``, api.py:25,63).
"""
from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch
import torch.distributed as dist

from .global_env import global_config

_INITIALIZED = False
_LOCAL_RANK = 0


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def local_rank() -> int:
    return _LOCAL_RANK


def device() -> torch.device:
    if torch.cuda.is_available():
        return torch.device("cuda", _LOCAL_RANK)
    return torch.device("cpu")


def init_distributed(backend: Optional[str] = None,
                     timeout_s: float = 600.0) -> None:
    """Initialize the torch.distributed world from environment variables.

    Launched via ``torch.distributed.run`` (one rank per GPU).  Falls back to
    a single-process (non-distributed) mode when RANK/WORLD_SIZE are absent,
    so single-GPU and CPU unit-test paths need no launcher.
    """
    global _INITIALIZED, _LOCAL_RANK
    if _INITIALIZED or is_distributed():
        _INITIALIZED = True
        return
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        # single-process mode
        _INITIALIZED = True
        if torch.cuda.is_available():
            torch.cuda.set_device(0)
        return
    backend = backend or global_config.resolved_dist_backend()
    _LOCAL_RANK = int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))
    if torch.cuda.is_available():
        torch.cuda.set_device(_LOCAL_RANK)
    dist.init_process_group(
        backend=backend,
        timeout=datetime.timedelta(seconds=timeout_s),
    )
    _INITIALIZED = True


def shutdown() -> None:
    global _INITIALIZED
    if is_distributed():
        dist.barrier()
        dist.destroy_process_group()
    _INITIALIZED = False


# ---------------------------------------------------------------------------
# Virtual (planning-time) mesh
# ---------------------------------------------------------------------------


@dataclass(frozen=True)
class VirtualMesh:
    """Unallocated mesh: a set of global ranks with a logical 2-D shape.

    Used by the shard/pipeline planners before any communicator exists.
    ``slice_ranks`` carves pipeline-stage submeshes out of it (analog of
    ``VirtualPhysicalMesh.slice_2d``, device_mesh.py:1888).
    """
    ranks: Tuple[int, ...]
    shape: Tuple[int, int]  # (dim0, dim1); dim0*dim1 == len(ranks)

    def __post_init__(self):
        assert self.shape[0] * self.shape[1] == len(self.ranks), \
            f"shape {self.shape} != #ranks {len(self.ranks)}"

    @property
    def num_devices(self) -> int:
        return len(self.ranks)

    def slice_ranks(self, start: int, num: int,
                    shape: Optional[Tuple[int, int]] = None) -> "VirtualMesh":
        sub = self.ranks[start:start + num]
        assert len(sub) == num
        if shape is None:
            shape = (1, num)
        return VirtualMesh(tuple(sub), shape)

    def reshape(self, shape: Tuple[int, int]) -> "VirtualMesh":
        return VirtualMesh(self.ranks, shape)

    def rank_grid(self) -> np.ndarray:
        return np.array(self.ranks).reshape(self.shape)


def full_virtual_mesh(num_devices: Optional[int] = None) -> VirtualMesh:
    n = num_devices if num_devices is not None else world_size()
    return VirtualMesh(tuple(range(n)), (1, n))


# ---------------------------------------------------------------------------
# Live mesh
# ---------------------------------------------------------------------------

# Cache process groups by the exact rank tuple so meshes sharing an axis reuse
# a single RCCL communicator (communicator creation is collective & expensive).
_GROUP_CACHE: Dict[Tuple[int, ...], dist.ProcessGroup] = {}


def _get_group(ranks: Sequence[int]) -> Optional[dist.ProcessGroup]:
    """Create (collectively, on all ranks!) or fetch a cached subgroup."""
    key = tuple(ranks)
    if key in _GROUP_CACHE:
        return _GROUP_CACHE[key]
    if not is_distributed():
        _GROUP_CACHE[key] = None
        return None
    if len(key) == world_size() and key == tuple(range(world_size())):
        g = dist.group.WORLD
    else:
        g = dist.new_group(ranks=list(key))
    _GROUP_CACHE[key] = g
    return g


class DeviceMesh:
    """A live logical 2-D mesh over torch.distributed ranks.

    Axis 0 is conventionally the data-parallel axis and axis 1 the
    tensor/model-parallel axis (matching the reference's logical mesh in
    ``auto_sharding.py:81``), but the solver is free to map either way.

    All ranks of the world must construct every mesh (group creation is a
    collective).  Ranks not in ``ranks`` hold the object for planning but have
    ``is_member == False``.
    """

    def __init__(self, ranks: Sequence[int], shape: Tuple[int, int]):
        assert shape[0] * shape[1] == len(ranks)
        self.ranks: Tuple[int, ...] = tuple(ranks)
        self.shape = shape
        self.grid = np.array(self.ranks).reshape(shape)
        self._my_rank = rank()
        self.is_member = self._my_rank in self.ranks

        # full-mesh group
        self.group = _get_group(self.ranks)
        # per-axis groups: for each row -> group over that row's ranks, etc.
        self._axis_groups: List[Optional[dist.ProcessGroup]] = [None, None]
        self._axis_ranks: List[Tuple[int, ...]] = [(), ()]
        # groups along axis 0 (varying dim0, fixed dim1 coordinate) — "columns"
        for col in range(shape[1]):
            col_ranks = tuple(self.grid[:, col].tolist())
            g = _get_group(col_ranks)
            if self._my_rank in col_ranks:
                self._axis_groups[0] = g
                self._axis_ranks[0] = col_ranks
        # groups along axis 1 — "rows"
        for row in range(shape[0]):
            row_ranks = tuple(self.grid[row, :].tolist())
            g = _get_group(row_ranks)
            if self._my_rank in row_ranks:
                self._axis_groups[1] = g
                self._axis_ranks[1] = row_ranks

        if self.is_member:
            pos = np.argwhere(self.grid == self._my_rank)[0]
            self.coord: Tuple[int, int] = (int(pos[0]), int(pos[1]))
        else:
            self.coord = (-1, -1)

        # Dedicated comm stream for overlap (HIP stream; RCCL launches there).
        self._comm_stream: Optional[torch.cuda.Stream] = None
        if torch.cuda.is_available():
            self._comm_stream = torch.cuda.Stream()

    # -------------------- topology queries --------------------
    @property
    def num_devices(self) -> int:
        return len(self.ranks)

    def axis_size(self, axis: int) -> int:
        return self.shape[axis]

    def axis_group(self, axis: int) -> Optional[dist.ProcessGroup]:
        return self._axis_groups[axis]

    def axis_ranks(self, axis: int) -> Tuple[int, ...]:
        return self._axis_ranks[axis]

    def axis_index(self, axis: int) -> int:
        """This rank's coordinate along `axis`."""
        return self.coord[axis]

    @property
    def comm_stream(self) -> Optional[torch.cuda.Stream]:
        return self._comm_stream

    # -------------------- collectives --------------------
    # Thin wrappers: single-process worlds are no-ops so the same step program
    # runs everywhere. `axis=None` means the full mesh.

    def _group_for(self, axis: Optional[int]):
        if axis is None:
            return self.group, self.num_devices
        return self._axis_groups[axis], self.shape[axis]

    def all_reduce(self, t: torch.Tensor, axis: Optional[int] = None,
                   op=dist.ReduceOp.SUM, async_op: bool = False):
        g, n = self._group_for(axis)
        if n == 1 or not is_distributed():
            return None
        return dist.all_reduce(t, op=op, group=g, async_op=async_op)

    def all_gather(self, out: torch.Tensor, t: torch.Tensor,
                   axis: Optional[int] = None, async_op: bool = False):
        g, n = self._group_for(axis)
        if n == 1 or not is_distributed():
            out.copy_(t)
            return None
        return dist.all_gather_into_tensor(out, t, group=g, async_op=async_op)

    def reduce_scatter(self, out: torch.Tensor, t: torch.Tensor,
                       axis: Optional[int] = None, async_op: bool = False):
        g, n = self._group_for(axis)
        if n == 1 or not is_distributed():
            out.copy_(t)
            return None
        return dist.reduce_scatter_tensor(out, t, group=g, async_op=async_op)

    def all_to_all(self, out: torch.Tensor, t: torch.Tensor,
                   axis: Optional[int] = None, async_op: bool = False):
        g, n = self._group_for(axis)
        if n == 1 or not is_distributed():
            out.copy_(t)
            return None
        return dist.all_to_all_single(out, t, group=g, async_op=async_op)

    def broadcast(self, t: torch.Tensor, src_coord: int = 0,
                  axis: Optional[int] = None, async_op: bool = False):
        g, n = self._group_for(axis)
        if n == 1 or not is_distributed():
            return None
        if axis is None:
            src = self.ranks[src_coord]
        else:
            src = self._axis_ranks[axis][src_coord]
        return dist.broadcast(t, src=src, group=g, async_op=async_op)

    def barrier(self):
        if is_distributed() and self.group is not None:
            dist.barrier(group=self.group)

    def check_alive(self, timeout_s: float = 30.0) -> bool:
        """Liveness probe over the mesh (reference MeshHostWorker.check_alive + driver polling, device_mesh.py:616 / pipeshard_executable.py:417): a tiny async all-reduce that every member must answer within the timeout.  Returns False instead of raising when a peer is dead/hung; detection-only, recovery = restart + restore_checkpoint, the same policy as the reference."""
        if not is_distributed() or self.group is None:
            return True
        t = torch.ones(1, device=device()
                       if torch.cuda.is_available() else "cpu")
        try:
            work = dist.all_reduce(t, group=self.group, async_op=True)
            import datetime
            ok = work.wait(datetime.timedelta(seconds=timeout_s))
            if ok is False:
                return False
            return bool(abs(float(t.item()) - len(self.ranks)) < 0.5)
        except Exception:
            return False

    # -------------------- p2p (cross-mesh / pipeline) --------------------
    @staticmethod
    def send(t: torch.Tensor, dst_rank: int, tag: int = 0):
        if not is_distributed():
            raise RuntimeError("send() requires a distributed world")
        dist.send(t, dst=dst_rank, tag=tag)

    @staticmethod
    def recv(t: torch.Tensor, src_rank: int, tag: int = 0):
        if not is_distributed():
            raise RuntimeError("recv() requires a distributed world")
        dist.recv(t, src=src_rank, tag=tag)

    @staticmethod
    def batch_isend_irecv(ops: List[dist.P2POp]):
        return dist.batch_isend_irecv(ops) if ops else []

    def __repr__(self):
        return f"DeviceMesh(shape={self.shape}, ranks={self.ranks})"


_MESH_CACHE: Dict[Tuple[Tuple[int, ...], Tuple[int, int]], DeviceMesh] = {}


def get_device_mesh(ranks: Sequence[int], shape: Tuple[int, int]) -> DeviceMesh:
    key = (tuple(ranks), tuple(shape))
    if key not in _MESH_CACHE:
        _MESH_CACHE[key] = DeviceMesh(ranks, shape)
    return _MESH_CACHE[key]


def full_mesh(shape: Optional[Tuple[int, int]] = None) -> DeviceMesh:
    n = world_size()
    if shape is None:
        shape = (1, n)
    return get_device_mesh(tuple(range(n)), shape)


def memory_stats() -> Dict[str, float]:
    """This rank's device memory stats in GB (reference MeshHostWorker
    get_memory_stats / get_max_memory_allocated, device_mesh.py:255-270).
    CPU-only environments report zeros."""
    if not torch.cuda.is_available():
        return {"allocated_gb": 0.0, "max_allocated_gb": 0.0,
                "reserved_gb": 0.0, "total_gb": 0.0}
    free, total = torch.cuda.mem_get_info()
    return {
        "allocated_gb": torch.cuda.memory_allocated() / 1e9,
        "max_allocated_gb": torch.cuda.max_memory_allocated() / 1e9,
        "reserved_gb": torch.cuda.memory_reserved() / 1e9,
        "total_gb": total / 1e9,
    }


def reset_memory_stats() -> None:
    """Reset the peak-memory counter (reference reset_memory_stats,
    device_mesh.py:268)."""
    if torch.cuda.is_available():
        torch.cuda.reset_peak_memory_stats()
