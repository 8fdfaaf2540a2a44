"""Pipeline execution engine: static 1F1B/GPipe microbatch runtime over
RCCL p2p (xGMI) between per-stage submeshes.

MI355X-native replacement for the reference's instruction-list interpreter
(``runtime_emitter.py:258`` emits RUN/SEND/RECV/FREE per worker;
``pipeshard_executable.py:489`` interprets them).  Here every rank executes
the same statically-known control flow for its stage; cross-stage traffic
is fused send+recv pairs (one batch_isend_irecv per adjacent-stage
exchange, both sides posting simultaneously — the deadlock-freedom
discipline the reference gets from sorted resharding task order,
SURVEY.md §5.2).

Stage meshes are uniform (same (dp, tp) shape): the p2p peer of a rank is
the same-coordinate rank of the adjacent stage mesh, so activations cross
stages with zero resharding.  (Heterogeneous stage meshes go through the
cross-mesh resharding module.)
"""
from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from ..mesh import DeviceMesh, is_distributed, rank
from ..parallel.grad_sync import GradSynchronizer
from . import schedules


class PipelineEngine:
    """Runs one training (or inference) step for this rank's stage."""

    def __init__(self, stage_module: torch.nn.Module, stage_idx: int,
                 num_stages: int, stage_mesh: DeviceMesh,
                 prev_peer: Optional[int], next_peer: Optional[int],
                 num_microbatches: int, act_shape: Tuple[int, ...],
                 act_dtype: torch.dtype, schedule: str = "1f1b",
                 grad_sync: Optional[GradSynchronizer] = None,
                 loss_scale_ranks: Optional[List[int]] = None):
        self.stage = stage_module
        self.s = stage_idx
        self.P = num_stages
        self.mesh = stage_mesh
        self.prev_peer = prev_peer  # global rank or None
        self.next_peer = next_peer
        self.M = num_microbatches
        self.act_shape = act_shape
        self.act_dtype = act_dtype
        self.schedule_name = schedule
        self.grad_sync = grad_sync
        self.is_first = stage_idx == 0
        self.is_last = stage_idx == num_stages - 1
        self.device = next(stage_module.parameters()).device
        self._act_shape_fn = None  # lazy: microbatch -> act shape
        self.loss_src_rank = 0
        self.tied_comms = []  # [(process_group, local param)]
        # heterogeneous stage meshes (reference auto-search submeshes):
        # activations cross mismatched boundaries via tile resharding
        self.hetero = False
        self.stage_meshes = None
        self.stage_shapes = None
        #: optional per-stage activation layout override:
        #: fn(stage_idx, (dp, tp), act_rank) -> dim-partition tuple.
        #: Default is batch-dim sharding over dp with tp replicas; a
        #: stage whose boundary activation is FEATURE-sharded (e.g. a
        #: col-parallel last layer without gather, or sequence parallel)
        #: declares it here and the tile exchange reshards it
        #: (reference scatter-allgather rewrite,
        #: cross_mesh_resharding.py:995).
        self.boundary_parts_fn = None
        self._specs = None  # (prev_fwd, prev_bwd, next_fwd, next_bwd)

    # ------------------------- p2p primitives -------------------------
    def _p2p(self, ops: List[dist.P2POp]):
        if not ops:
            return
        works = dist.batch_isend_irecv(ops)
        for w in works:
            w.wait()

    # heterogeneous boundaries: tile-resharding exchange specs
    def _stage_parts(self, stage: int):
        dp, tp = self.stage_shapes[stage]
        if self.boundary_parts_fn is not None:
            p = self.boundary_parts_fn(stage, (dp, tp),
                                       len(self.act_shape))
            if p is not None:
                return tuple(p)
        return (dp,) + (1,) * (len(self.act_shape) - 1)

    def _act_placement(self, stage: int):
        import numpy as np
        from ..parallel.resharding import Placement
        dp, tp = self.stage_shapes[stage]
        grid = self.stage_meshes[stage].grid
        parts = self._stage_parts(stage)
        n_tiles = int(np.prod(parts))
        if parts[0] == dp and n_tiles == dp:
            # batch-only sharding, tp replicas: replica-major layout
            # (tile b replica t = grid[b, t])
            ranks = tuple(int(r) for r in grid.flatten(order="F"))
        elif n_tiles == dp * tp:
            # batch x feature tiling: row-major tiles (tile (b, t) =
            # grid[b, t])
            ranks = tuple(int(r) for r in grid.flatten(order="C"))
        elif n_tiles == tp:
            # feature-only sharding, dp replicas (tile t replica b =
            # grid[b, t])
            ranks = tuple(int(r) for r in grid.flatten(order="C"))
        else:
            raise NotImplementedError(
                f"boundary parts {parts} on stage mesh ({dp},{tp})")
        return Placement(tuple(self.act_shape), parts, ranks)

    def _build_specs(self):
        from ..global_env import global_config
        from ..parallel.resharding import ReshardingTaskSpec
        sa = global_config.use_local_allgather
        pf = pb = nf = nb = None
        if self.s > 0:
            a, b = self._act_placement(self.s - 1), self._act_placement(self.s)
            pf = ReshardingTaskSpec.build(a, b, scatter_allgather=sa)
            pb = ReshardingTaskSpec.build(b, a, scatter_allgather=sa)
        if self.s < self.P - 1:
            a, b = self._act_placement(self.s), self._act_placement(self.s + 1)
            nf = ReshardingTaskSpec.build(a, b, scatter_allgather=sa)
            nb = ReshardingTaskSpec.build(b, a, scatter_allgather=sa)
        self._specs = (pf, pb, nf, nb)

    def _local_tile_shape(self) -> tuple:
        parts = self._stage_parts(self.s)
        return tuple(d // p for d, p in zip(self.act_shape, parts))

    def _hetero_exchange(self, send=None, recv_prev=False, recv_next=False,
                         send_prev=False, send_next=False):
        """One batched exchange combining up to one send and one recv with
        the adjacent stages (both sides post their batch at the matching
        schedule slot — deadlock-free)."""
        from ..parallel.resharding import apply_fixups, prepare_resharding
        if self._specs is None:
            self._build_specs()
        pf, pb, nf, nb = self._specs
        ops, fixups = [], []
        buf = None
        if send_next:  # forward activation to next stage
            o, f = prepare_resharding(nf, send.contiguous(), None)
            ops += o; fixups += f
        if send_prev:  # gradient to previous stage
            o, f = prepare_resharding(pb, send.contiguous(), None)
            ops += o; fixups += f
        if recv_prev:  # activation from previous stage
            buf = torch.empty(self._local_tile_shape(), dtype=self.act_dtype,
                              device=self.device)
            o, f = prepare_resharding(pf, None, buf)
            ops += o; fixups += f
        if recv_next:  # gradient from next stage
            buf = torch.empty(self._local_tile_shape(), dtype=self.act_dtype,
                              device=self.device)
            o, f = prepare_resharding(nb, None, buf)
            ops += o; fixups += f
        self._p2p(ops)
        apply_fixups(fixups)
        if buf is not None:
            from ..parallel.resharding import apply_allgather_fixes
            apply_allgather_fixes(pf if recv_prev else nb, buf)
        return buf

    def _recv_forward(self) -> torch.Tensor:
        if self.hetero:
            return self._hetero_exchange(recv_prev=True)
        buf = torch.empty(self.act_shape, dtype=self.act_dtype,
                          device=self.device)
        self._p2p([dist.P2POp(dist.irecv, buf, self.prev_peer)])
        return buf

    def _send_forward(self, y: torch.Tensor):
        if self.hetero:
            self._hetero_exchange(send=y, send_next=True)
            return
        self._p2p([dist.P2POp(dist.isend, y.contiguous(), self.next_peer)])

    def _recv_backward(self) -> torch.Tensor:
        if self.hetero:
            return self._hetero_exchange(recv_next=True)
        buf = torch.empty(self.act_shape, dtype=self.act_dtype,
                          device=self.device)
        self._p2p([dist.P2POp(dist.irecv, buf, self.next_peer)])
        return buf

    def _send_backward(self, g: torch.Tensor):
        if self.hetero:
            self._hetero_exchange(send=g, send_prev=True)
            return
        self._p2p([dist.P2POp(dist.isend, g.contiguous(), self.prev_peer)])

    def _send_forward_recv_backward(self, y: torch.Tensor) -> torch.Tensor:
        if self.hetero:
            return self._hetero_exchange(send=y, send_next=True,
                                         recv_next=True)
        buf = torch.empty(self.act_shape, dtype=self.act_dtype,
                          device=self.device)
        self._p2p([dist.P2POp(dist.isend, y.contiguous(), self.next_peer),
                   dist.P2POp(dist.irecv, buf, self.next_peer)])
        return buf

    def _send_backward_recv_forward(self, g: torch.Tensor) -> torch.Tensor:
        if self.hetero:
            return self._hetero_exchange(send=g, send_prev=True,
                                         recv_prev=True)
        buf = torch.empty(self.act_shape, dtype=self.act_dtype,
                          device=self.device)
        self._p2p([dist.P2POp(dist.isend, g.contiguous(), self.prev_peer),
                   dist.P2POp(dist.irecv, buf, self.prev_peer)])
        return buf

    # ------------------------- tracing -------------------------
    def enable_tracing(self, tracer=None):
        """Record per-(microbatch x fwd/bwd) spans; dump with
        dump_stage_execution_trace (reference
        dump_stage_execution_trace_internal, pipeshard_executable.py:592).
        """
        from ..timer import Tracer
        self.tracer = tracer or Tracer()
        return self.tracer

    def dump_stage_execution_trace(self, path: str):
        """Write the recorded schedule execution as chrome://tracing JSON
        (one row per stage; fwd/bwd spans labeled by microbatch)."""
        assert getattr(self, "tracer", None) is not None,             "call enable_tracing() before the traced step"
        self.tracer.dump_chrome_trace(path, pid=self.s)

    def _trace(self, kind, mb_idx):
        import time as _t

        class _Span:

            def __init__(self, eng):
                self.eng = eng

            def __enter__(self):
                self.t0 = _t.perf_counter()

            def __exit__(self, *a):
                tr = getattr(self.eng, "tracer", None)
                if tr is not None:
                    tr.log_span(f"stage{self.eng.s}.{kind}.mb{mb_idx}",
                                self.t0, _t.perf_counter(), cat=kind,
                                tid=self.eng.s)

        return _Span(self)

    # ------------------------- compute steps -------------------------
    def _forward_step(self, mb_idx: int, x: Optional[torch.Tensor],
                      microbatches: List[Any]):
        """Returns (input, output): output is loss on the last stage."""
        with self._trace("fwd", mb_idx):
            return self._forward_step_inner(mb_idx, x, microbatches)

    def _forward_step_inner(self, mb_idx, x, microbatches):
        if not self.is_first:
            x = x.requires_grad_(True)
        mb = microbatches[mb_idx]
        if self.hetero:
            from ..data_loader import shard_batch
            dp = self.stage_shapes[self.s][0]
            mb = shard_batch(mb, dp, max(self.mesh.axis_index(0), 0))
        out = self.stage(x, mb)
        return x, out

    def _backward_step(self, x, out, out_grad,
                       is_last_bwd: bool) -> Optional[torch.Tensor]:
        with self._trace("bwd", "x"):
            return self._backward_step_inner(x, out, out_grad, is_last_bwd)

    def _backward_step_inner(self, x, out, out_grad,
                             is_last_bwd: bool) -> Optional[torch.Tensor]:
        if self.grad_sync is not None:
            self.grad_sync.begin_microbatch(is_last=is_last_bwd)
        if self.is_last:
            out.backward()  # out is the microbatch loss
        else:
            torch.autograd.backward(out, grad_tensors=out_grad)
        return None if self.is_first else x.grad

    # ------------------------- the step -------------------------
    def train_step(self, microbatches: List[Any]) -> torch.Tensor:
        """1F1B (or GPipe) over the local microbatch list.  Returns the mean
        loss (valid on last-stage ranks; others get the broadcast value via
        PipelineTrainer)."""
        assert len(microbatches) == self.M
        if self.act_shape is None:
            self.act_shape = tuple(self._act_shape_fn(microbatches[0]))
        P, M, s = self.P, self.M, self.s
        if self.grad_sync is not None:
            self.grad_sync.zero_grads()
            self.grad_sync.begin_microbatch(is_last=False)

        if self.schedule_name == "gpipe":
            return self._gpipe_step(microbatches)

        warmup = min(P - s - 1, M)
        steady = M - warmup
        fwd_i = 0
        bwd_i = 0
        pending: List[Tuple[Optional[torch.Tensor], torch.Tensor]] = []
        losses = []

        # ---- warmup forwards ----
        for _ in range(warmup):
            x = None if self.is_first else self._recv_forward()
            x, out = self._forward_step(fwd_i, x, microbatches)
            if not self.is_last:
                self._send_forward(out)
            else:
                losses.append(out.detach())
            pending.append((x, out))
            fwd_i += 1

        x_next = None
        if steady > 0 and not self.is_first:
            x_next = self._recv_forward()

        # ---- steady 1F1B ----
        for i in range(steady):
            x, out = self._forward_step(fwd_i, x_next, microbatches)
            fwd_i += 1
            if self.is_last:
                losses.append(out.detach())
                out_grad = None
            else:
                out_grad = self._send_forward_recv_backward(out)
            pending.append((x, out))
            px, pout = pending.pop(0)
            bwd_i += 1
            in_grad = self._backward_step(px, pout, out_grad,
                                          is_last_bwd=(bwd_i == M))
            last_iter = (i == steady - 1)
            if self.is_first:
                pass
            elif last_iter:
                self._send_backward(in_grad)
            else:
                x_next = self._send_backward_recv_forward(in_grad)

        # ---- cooldown backwards ----
        for _ in range(warmup):
            out_grad = None if self.is_last else self._recv_backward()
            px, pout = pending.pop(0)
            bwd_i += 1
            in_grad = self._backward_step(px, pout, out_grad,
                                          is_last_bwd=(bwd_i == M))
            if not self.is_first:
                self._send_backward(in_grad)

        if self.grad_sync is not None:
            self.grad_sync.finish()
        self._sync_tied_grads()
        if losses:
            return self._mean_loss(losses)
        return torch.zeros((), device=self.device)

    def _mean_loss(self, losses):
        loss = torch.stack(losses).mean()
        dp = self.mesh.axis_size(0)
        if self.is_last and dp > 1:
            loss = loss.clone()
            self.mesh.all_reduce(loss, axis=0)
            loss = loss / dp
        return loss

    def _sync_tied_grads(self):
        """Cross-stage all-reduce of tied-weight grads (reference
        cross-mesh allreduce for shared embeddings)."""
        if not self.tied_comms or not is_distributed():
            return
        for g, p in self.tied_comms:
            if p.grad is not None:
                dist.all_reduce(p.grad, group=g)

    def _gpipe_step(self, microbatches) -> torch.Tensor:
        M = self.M
        pending = []
        losses = []
        for i in range(M):
            x = None if self.is_first else self._recv_forward()
            x, out = self._forward_step(i, x, microbatches)
            if not self.is_last:
                self._send_forward(out)
            else:
                losses.append(out.detach())
            pending.append((x, out))
        for i in reversed(range(M)):
            out_grad = None if self.is_last else self._recv_backward()
            px, pout = pending[i]
            in_grad = self._backward_step(px, pout, out_grad,
                                          is_last_bwd=(i == 0))
            if not self.is_first:
                self._send_backward(in_grad)
        pending.clear()
        if self.grad_sync is not None:
            self.grad_sync.finish()
        self._sync_tied_grads()
        if losses:
            return self._mean_loss(losses)
        return torch.zeros((), device=self.device)

    # GPipe backward consumes grads in reverse mb order — receive order from
    # the next stage is also reverse, consistent across stages.

    def inference_step(self, microbatches: List[Any]) -> List[torch.Tensor]:
        """Forward-only pipeline; returns last-stage outputs per microbatch."""
        outs = []
        if self.act_shape is None:
            self.act_shape = tuple(self._act_shape_fn(microbatches[0]))
        with torch.no_grad():
            for i in range(len(microbatches)):
                x = None if self.is_first else self._recv_forward()
                out = self.stage(x, microbatches[i])
                if not self.is_last:
                    self._send_forward(out)
                else:
                    outs.append(out)
        return outs
