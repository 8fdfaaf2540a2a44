"""Bucketed gradient synchronization over the data-parallel mesh axis.

MI355X-native replacement for the reference's grad-sync machinery: the SPMD
partitioner's grad all-reduce + thresholded combiner
plus the skip-grad-sync gating used during microbatch accumulation
(``mesh_executable.py:886-896`` runs accumulate_grad with the grad all-reduce
channels disabled until the last microbatch).

Design:
- Grads live in *flat per-bucket buffers*; each param's ``.grad`` is a view
  into its bucket, so backward accumulates in place with zero extra copies.
- Buckets are ordered reverse-registration ≈ backward completion order and
  sized for xGMI: each ring all-reduce is bound by one ~153 GB/s link, so
  buckets default to 100 MiB to amortize latency (global_config).
- When the last grad of a bucket lands (post-accumulate-grad hook) *and*
  sync is enabled (last microbatch), the all-reduce (or reduce-scatter for
  ZeRO-2) launches asynchronously on the mesh comm stream, overlapping the
  remaining backward compute.
- ``sync_enabled=False`` (earlier microbatches) makes the hook a no-op —
  the skip-collective gate.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

import torch
import torch.distributed as dist

from ..global_env import global_config
from ..mesh import DeviceMesh, is_distributed


@dataclass
class _Bucket:
    params: List[torch.nn.Parameter] = field(default_factory=list)
    numel: int = 0
    flat: Optional[torch.Tensor] = None
    #: expert-parallel params: grads complete locally, no dp collective
    no_sync: bool = False
    # per-step state
    pending: int = 0
    work: Optional[object] = None  # dist.Work
    comm_event: Optional[torch.cuda.Event] = None
    # zero2: this rank's shard view of `flat`
    shard: Optional[torch.Tensor] = None


class GradSynchronizer:
    """Owns flat grad storage + overlapped collectives for one param set."""

    def __init__(self, params: List[torch.nn.Parameter],
                 mesh: Optional[DeviceMesh], axis: int = 0,
                 bucket_bytes: Optional[int] = None,
                 reduce_scatter: bool = False,
                 grad_dtype: Optional[torch.dtype] = None):
        self.mesh = mesh
        self.axis = axis
        self.dp = mesh.axis_size(axis) if mesh is not None else 1
        self.reduce_scatter = reduce_scatter and self.dp > 1
        self.sync_enabled = True
        bucket_bytes = bucket_bytes or global_config.grad_bucket_bytes

        params = [p for p in params if p.requires_grad]
        self.params = params
        # reverse order ≈ backward completion order; expert-parallel params
        # (marked by ExpertParallelMLP) go to dedicated no-sync buckets
        order = list(reversed(params))
        self.buckets: List[_Bucket] = []
        cur = _Bucket()
        cur_local = _Bucket(no_sync=True)
        for p in order:
            tgt = cur_local if getattr(p, "_expert_parallel", False) else cur
            esize = p.element_size()
            if tgt.params and (tgt.numel + p.numel()) * esize > bucket_bytes:
                self.buckets.append(tgt)
                if tgt is cur:
                    cur = tgt = _Bucket()
                else:
                    cur_local = tgt = _Bucket(no_sync=True)
            tgt.params.append(p)
            tgt.numel += p.numel()
        if cur.params:
            self.buckets.append(cur)
        if cur_local.params:
            self.buckets.append(cur_local)

        # allocate flat buffers + wire .grad views
        self._param_bucket = {}
        for b in self.buckets:
            # pad so reduce-scatter shards evenly
            pad = (-b.numel) % (self.dp if self.reduce_scatter else 1)
            total = b.numel + pad
            dev = b.params[0].device
            dt = grad_dtype or b.params[0].dtype
            b.flat = torch.zeros(total, dtype=dt, device=dev)
            off = 0
            for p in b.params:
                n = p.numel()
                p.grad = b.flat[off:off + n].view(p.shape)
                off += n
                self._param_bucket[p] = b
            if self.reduce_scatter:
                shard_n = total // self.dp
                r = mesh.axis_index(axis)
                b.shard = b.flat[r * shard_n:(r + 1) * shard_n]

        # install hooks
        self._hooks = []
        for p in params:
            h = p.register_post_accumulate_grad_hook(self._on_grad)
            self._hooks.append(h)
        self._reset_pending()

    # ------------------------------------------------------------------
    def _reset_pending(self):
        for b in self.buckets:
            b.pending = len(b.params)
            b.work = None
            b.comm_event = None

    def _on_grad(self, p: torch.nn.Parameter):
        if not self.sync_enabled:
            return
        b = self._param_bucket[p]
        b.pending -= 1
        if b.pending == 0:
            self._launch(b)

    def _launch(self, b: _Bucket):
        if b.no_sync or self.dp == 1 or not is_distributed():
            return
        group = self.mesh.axis_group(self.axis)
        from ..global_env import global_config
        stream = self.mesh.comm_stream \
            if global_config.overlap_grad_sync else None
        if stream is not None:
            # comm stream waits for the producing compute work
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                if self.reduce_scatter:
                    dist.reduce_scatter_tensor(b.shard, b.flat, group=group)
                else:
                    dist.all_reduce(b.flat, group=group)
                b.comm_event = torch.cuda.Event()
                b.comm_event.record(stream)
        else:
            if self.reduce_scatter:
                b.work = dist.reduce_scatter_tensor(b.shard, b.flat,
                                                    group=group, async_op=True)
            else:
                b.work = dist.all_reduce(b.flat, group=group, async_op=True)

    # ------------------------------------------------------------------
    def begin_microbatch(self, is_last: bool):
        """Call before each microbatch backward: enables sync only on the
        last microbatch (grad-accumulation skip gate, reference N5)."""
        self.sync_enabled = is_last
        if is_last:
            self._reset_pending()

    def finish(self):
        """Wait for all in-flight collectives (end of backward)."""
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
                b.work = None
            if b.comm_event is not None:
                torch.cuda.current_stream().wait_event(b.comm_event)
                b.comm_event = None
        # straggler buckets whose hooks never completed a launch (e.g. params
        # without grads this step) are synced here
        if self.sync_enabled and self.dp > 1 and is_distributed():
            for b in self.buckets:
                if b.pending > 0:
                    self._launch(b)
                    b.pending = 0
            for b in self.buckets:
                if b.work is not None:
                    b.work.wait()
                    b.work = None
                if b.comm_event is not None:
                    torch.cuda.current_stream().wait_event(b.comm_event)
                    b.comm_event = None

    def zero_grads(self):
        for b in self.buckets:
            b.flat.zero_()
        self._reset_pending()

    # ------- ZeRO-2 helpers -------
    def shard_views(self):
        """(flat_shard_grad, [(param, param_flat_slice_in_shard)...]) per
        bucket — used by the sharded optimizer."""
        assert self.reduce_scatter
        return [b.shard for b in self.buckets]

    def allgather_params(self):
        """ZeRO-2: after the sharded update wrote this rank's slice of each
        bucket-image param buffer, all-gather the full params."""
        raise NotImplementedError  # installed by zero.py

    def detach_hooks(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []
