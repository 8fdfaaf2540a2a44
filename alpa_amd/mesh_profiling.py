"""Op/collective profiler + persistent cost database.

Capability analog of the reference's ``alpa/mesh_profiling.py``:
MeshProfilingResult:18 (cost curves + interpolating estimator),
ProfilingResultDatabase:162 (keyed (cluster_key, mesh_shape), pickle
save/load), profile_one_hlo_op:392, profile_all:725,
estimate_hlo_module_cost:901.

On MI355X the measurements come from hipEvent-timed torch ops (matmul via
hipBLASLt, our HIP kernels, RCCL collectives through the mesh); the
alpha-beta mesh model (shard_parallel/ir.MeshModel) and the stage DP are
recalibrated from these numbers.
"""
from __future__ import annotations

import bisect
import pickle
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

import torch

from .mesh import DeviceMesh


def benchmark_func(fn: Callable, warmup: int = 2, repeat: int = 5,
                   sync: bool = True) -> float:
    """Median wall time of fn() (reference benchmark_func, util.py:1053)."""
    for _ in range(warmup):
        fn()
    if sync and torch.cuda.is_available():
        torch.cuda.synchronize()
    times = []
    for _ in range(repeat):
        t0 = time.perf_counter()
        fn()
        if sync and torch.cuda.is_available():
            torch.cuda.synchronize()
        times.append(time.perf_counter() - t0)
    times.sort()
    return times[len(times) // 2]


class CostCurve:
    """Monotonic size->time curve with log-linear interpolation
    (reference MeshProfilingResult._estimate_internal:109 + monotonic
    clean-up :44)."""

    def __init__(self):
        self.sizes: List[float] = []
        self.times: List[float] = []

    def add(self, size: float, t: float):
        i = bisect.bisect_left(self.sizes, size)
        self.sizes.insert(i, size)
        self.times.insert(i, t)
        # enforce monotonicity
        for j in range(i + 1, len(self.times)):
            self.times[j] = max(self.times[j], self.times[j - 1])

    def estimate(self, size: float) -> float:
        if not self.sizes:
            raise ValueError("empty cost curve")
        if size <= self.sizes[0]:
            return self.times[0]
        if size >= self.sizes[-1]:
            # extrapolate with the last slope
            if len(self.sizes) >= 2:
                s0, s1 = self.sizes[-2], self.sizes[-1]
                t0, t1 = self.times[-2], self.times[-1]
                if s1 > s0:
                    return t1 + (size - s1) * (t1 - t0) / (s1 - s0)
            return self.times[-1]
        i = bisect.bisect_left(self.sizes, size)
        s0, s1 = self.sizes[i - 1], self.sizes[i]
        t0, t1 = self.times[i - 1], self.times[i]
        w = (size - s0) / (s1 - s0)
        return t0 * (1 - w) + t1 * w


@dataclass
class MeshProfilingResult:
    """Per-(mesh shape) measured cost curves."""
    mesh_shape: Tuple[int, int] = (1, 1)
    #: op name ("matmul_bf16", "layer_norm", ...) -> flops-or-bytes curve
    op_curves: Dict[str, CostCurve] = field(default_factory=dict)
    #: collective ("all_reduce", axis) -> bytes curve
    coll_curves: Dict[Tuple[str, int], CostCurve] = field(
        default_factory=dict)
    #: measured scalar facts, e.g. "gpt_act_bytes_per_token_hidden" —
    #: activation bytes a transformer block pins for backward, per
    #: token per hidden unit (tools/measure_memory.py); read via
    #: getattr(r, "scalars", {}) so pre-existing pickles stay loadable
    scalars: Dict[str, float] = field(default_factory=dict)

    def estimate_op(self, op: str, size: float) -> float:
        return self.op_curves[op].estimate(size)

    def estimate_collective(self, kind: str, axis: int,
                            bytes_: float) -> float:
        return self.coll_curves[(kind, axis)].estimate(bytes_)


class ProfilingResultDatabase:
    """(cluster_key, mesh_shape) -> MeshProfilingResult, pickled
    (reference ProfilingResultDatabase:162)."""

    def __init__(self):
        self.data: Dict[Tuple[str, Tuple[int, int]], MeshProfilingResult] = {}

    def update_one_mesh(self, cluster_key: str, mesh_shape: Tuple[int, int],
                        result: MeshProfilingResult):
        self.data[(cluster_key, tuple(mesh_shape))] = result

    def query(self, cluster_key: str,
              mesh_shape: Tuple[int, int]) -> MeshProfilingResult:
        return self.data[(cluster_key, tuple(mesh_shape))]

    def insert_dummy_mesh_result(self, cluster_key: str,
                                 mesh_shape: Tuple[int, int]):
        """Analytic fallback entry so device-free planners run without
        profiling (reference insert_dummy_mesh_result:185)."""
        r = MeshProfilingResult(mesh_shape)
        c = CostCurve()
        # roofline-ish: 1.2 PF matmul
        for f in (1e9, 1e11, 1e13):
            c.add(f, f / 1.2e15 + 5e-6)
        r.op_curves["matmul_bf16"] = c
        for kind in ("all_reduce", "all_gather", "reduce_scatter",
                     "all_to_all"):
            for axis in (0, 1):
                cc = CostCurve()
                n = mesh_shape[axis]
                for b in (1e5, 1e7, 1e9):
                    factor = 2 * (n - 1) / n if kind == "all_reduce" \
                        else (n - 1) / n
                    cc.add(b, 1e-5 + factor * b / 150e9)
                r.coll_curves[(kind, axis)] = cc
        self.update_one_mesh(cluster_key, mesh_shape, r)

    def save(self, path: str):
        with open(path, "wb") as f:
            pickle.dump(self.data, f)

    def load(self, path: str):
        with open(path, "rb") as f:
            self.data.update(pickle.load(f))


def profile_matmul(sizes=((2048, 2048, 2048), (4096, 4096, 4096),
                          (8192, 8192, 8192)),
                   dtype=torch.bfloat16, device="cuda") -> CostCurve:
    """Measured hipBLASLt GEMM curve (flops -> seconds)."""
    c = CostCurve()
    for (m, k, n) in sizes:
        a = torch.randn(m, k, dtype=dtype, device=device)
        b = torch.randn(k, n, dtype=dtype, device=device)
        t = benchmark_func(lambda: a @ b)
        c.add(2.0 * m * k * n, t)
    return c


def profile_collective(mesh: DeviceMesh, kind: str, axis: int,
                       sizes=(1 << 20, 1 << 24, 1 << 27),
                       device="cuda") -> CostCurve:
    """Measured RCCL collective curve over one mesh axis (bytes -> s)."""
    import torch.distributed as dist
    c = CostCurve()
    n = mesh.axis_size(axis)
    for nbytes in sizes:
        numel = nbytes // 2
        t_in = torch.randn(numel, dtype=torch.bfloat16, device=device)
        if kind == "all_reduce":
            fn = lambda: mesh.all_reduce(t_in, axis=axis)
        elif kind == "all_gather":
            out = torch.empty(numel * n, dtype=torch.bfloat16, device=device)
            fn = lambda: mesh.all_gather(out, t_in, axis=axis)
        elif kind == "reduce_scatter":
            outs = torch.empty(numel // n, dtype=torch.bfloat16,
                               device=device)
            fn = lambda: mesh.reduce_scatter(outs, t_in, axis=axis)
        elif kind == "all_to_all":
            out = torch.empty_like(t_in)
            fn = lambda: mesh.all_to_all(out, t_in, axis=axis)
        else:
            raise ValueError(kind)
        c.add(float(nbytes), benchmark_func(fn))
    return c


def profile_all(mesh: Optional[DeviceMesh], cluster_key: str = "mi355x",
                device="cuda") -> ProfilingResultDatabase:
    """Profile ops (+ collectives when distributed) into a database
    (reference profile_all:725)."""
    db = ProfilingResultDatabase()
    shape = mesh.shape if mesh is not None else (1, 1)
    r = MeshProfilingResult(shape)
    r.op_curves["matmul_bf16"] = profile_matmul(device=device)
    if mesh is not None and mesh.num_devices > 1:
        for kind in ("all_reduce", "all_gather", "reduce_scatter",
                     "all_to_all"):
            for axis in (0, 1):
                if mesh.axis_size(axis) > 1:
                    r.coll_curves[(kind, axis)] = profile_collective(
                        mesh, kind, axis, device=device)
    db.update_one_mesh(cluster_key, shape, r)
    return db


def estimate_stage_cost(db: ProfilingResultDatabase, cluster_key: str,
                        mesh_shape: Tuple[int, int], matmul_flops: float,
                        collective_bytes: Dict[Tuple[str, int], float]
                        ) -> float:
    """Stage latency estimate from the profiled DB (reference
    estimate_hlo_module_cost:901) — feeds the stage-construction DP's
    cost-model path."""
    r = db.query(cluster_key, mesh_shape)
    t = r.estimate_op("matmul_bf16", matmul_flops)
    for (kind, axis), b in collective_bytes.items():
        t += r.estimate_collective(kind, axis, b)
    return t
