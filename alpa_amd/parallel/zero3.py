"""ZeRO-3: parameters sharded across dp, all-gathered just-in-time around
each module's forward/backward and released after use.

Reference: Zero3Parallel (parallel_method.py:146) = force_zero_stage_3 +
all-gather threshold (auto_sharding.py:225-236) — weights live sharded and
the partitioner inserts all-gathers around use.  Here the unit of
gathering is a module group (e.g. one transformer block): its params are
flattened into one bucket; between uses each rank stores only its
1/dp shard (+ sharded fp32 moments), so steady-state memory is
params/dp + activations.

Flow per step:
  fwd:  pre-forward hook all-gathers the block's bucket (params become
        views of the gathered flat); post-forward releases it
  bwd:  full-backward pre-hook re-gathers; after the block's grads are
        complete, grads are reduce-scattered to shards and the gathered
        params + full grads are freed
  step: fused AdamW on (param shard, grad shard, fp32 moment shards),
        no gathering needed.
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from .. import ops
from ..mesh import DeviceMesh, is_distributed


class _BlockShard:
    """One module group's flattened parameters with FSDP-style storage
    management: `full` (the gathered flat buffer) is allocated once and its
    storage resized to 0 between uses — param views and autograd-saved
    tensors stay pointed at the same storage, so re-gathering before
    backward revalidates them (the standard ZeRO-3/FSDP trick)."""

    def __init__(self, module: nn.Module, dp: int, rank_idx: int):
        self.module = module
        self.rank_idx = rank_idx
        self.params = [p for p in module.parameters() if p.requires_grad]
        self.shapes = [p.shape for p in self.params]
        self.numels = [p.numel() for p in self.params]
        numel = sum(self.numels)
        self.pad = (-numel) % dp
        self.total = numel + self.pad
        self.shard_n = self.total // dp
        dev = self.params[0].device
        dt = self.params[0].dtype
        self.full = torch.zeros(self.total, dtype=dt, device=dev)
        off = 0
        self.offsets = []
        with torch.no_grad():
            for p, n in zip(self.params, self.numels):
                self.full[off:off + n].copy_(p.data.reshape(-1))
                self.offsets.append(off)
                off += n
        self.shard = self.full[rank_idx * self.shard_n:
                               (rank_idx + 1) * self.shard_n].clone()
        # params become permanent views of `full`
        with torch.no_grad():
            for p, offn, shape, n in zip(self.params, self.offsets,
                                         self.shapes, self.numels):
                p.data = self.full[offn:offn + n].view(shape)
        self.grad_flat = torch.zeros(self.total, dtype=dt, device=dev)
        for p, offn, shape, n in zip(self.params, self.offsets, self.shapes,
                                     self.numels):
            p.grad = self.grad_flat[offn:offn + n].view(shape)
        self.grad_shard = torch.zeros_like(self.shard)
        self._nbytes = self.full.untyped_storage().nbytes()
        self._gbytes = self.grad_flat.untyped_storage().nbytes()
        self.release()
        self.release_grads()

    @property
    def gathered(self) -> bool:
        return self.full.untyped_storage().nbytes() > 0

    def gather(self, group):
        if self.gathered:
            return
        self.full.untyped_storage().resize_(self._nbytes)
        if is_distributed() and group is not None:
            dist.all_gather_into_tensor(self.full, self.shard, group=group)
        else:
            self.full.copy_(self.shard)

    def release(self):
        self.full.untyped_storage().resize_(0)

    def writeback_shard(self):
        """Copy this rank's slice of the (gathered) full buffer back into
        the authoritative shard — needed after external writes into the
        param views (checkpoint restore)."""
        with torch.no_grad():
            self.shard.copy_(self.full[self.rank_idx * self.shard_n:
                                       (self.rank_idx + 1) * self.shard_n])

    def prepare_grads(self):
        if self.grad_flat.untyped_storage().nbytes() == 0:
            self.grad_flat.untyped_storage().resize_(self._gbytes)
        self.grad_flat.zero_()

    def reduce_grads(self, group):
        """reduce-scatter this microbatch's grads and ACCUMULATE into the
        shard (grad accumulation across microbatches)."""
        tmp = torch.empty_like(self.grad_shard)
        if is_distributed() and group is not None:
            dist.reduce_scatter_tensor(tmp, self.grad_flat, group=group)
        else:
            tmp.copy_(self.grad_flat)
        self.grad_shard.add_(tmp)
        self.release_grads()

    def release_grads(self):
        self.grad_flat.untyped_storage().resize_(0)


class Zero3Manager:
    """Installs gather/release hooks on the given module groups."""

    def __init__(self, blocks: List[nn.Module], mesh: Optional[DeviceMesh],
                 axis: int = 0):
        self.mesh, self.axis = mesh, axis
        self.dp = mesh.axis_size(axis) if mesh is not None else 1
        self.group = mesh.axis_group(axis) if mesh is not None else None
        idx = mesh.axis_index(axis) if (mesh is not None and mesh.is_member) \
            else 0
        self.blocks = [_BlockShard(m, max(self.dp, 1), max(idx, 0))
                       for m in blocks]
        self._by_module = {b.module: b for b in self.blocks}
        self._by_param = {}
        self._grads_ready: Dict[int, int] = {}
        for b in self.blocks:
            b.module.register_forward_pre_hook(self._pre_fwd)
            b.module.register_forward_hook(self._post_fwd)
            b.module.register_full_backward_pre_hook(self._pre_bwd)
            for p in b.params:
                self._by_param[id(p)] = b
                p.register_post_accumulate_grad_hook(self._on_grad)

    # -------------------- hooks --------------------
    def _pre_fwd(self, module, inputs):
        self._by_module[module].gather(self.group)

    def _post_fwd(self, module, inputs, output):
        # release between fwd and bwd (storage-resize keeps views valid)
        self._by_module[module].release()

    def _pre_bwd(self, module, grad_output):
        b = self._by_module[module]
        b.gather(self.group)
        b.prepare_grads()
        self._grads_ready[id(b)] = len(b.params)

    def _on_grad(self, p):
        b = self._by_param[id(p)]
        self._grads_ready[id(b)] -= 1
        if self._grads_ready[id(b)] == 0:
            b.reduce_grads(self.group)
            b.release()

    # -------------------- optimizer view --------------------
    def shards(self):
        return ([b.shard for b in self.blocks],
                [b.grad_shard for b in self.blocks])

    def zero_grads(self):
        for b in self.blocks:
            b.grad_shard.zero_()

    def materialized(self):
        """Context manager: gather every block's params (so external code
        like checkpoint save/restore can read/write them), write the
        shards back and release on exit."""
        import contextlib

        @contextlib.contextmanager
        def cm():
            for b in self.blocks:
                b.gather(self.group)
            try:
                yield
            finally:
                for b in self.blocks:
                    b.writeback_shard()
                    b.release()

        return cm()

    def gathered_bytes(self) -> int:
        """Currently-resident gathered bytes (memory assertion hook)."""
        return sum(b.full.untyped_storage().nbytes() for b in self.blocks)


class Zero3Optimizer:
    """Fused AdamW over the ZeRO-3 shard set."""

    def __init__(self, manager: Zero3Manager, lr: float = 1e-4,
                 betas=(0.9, 0.95), eps: float = 1e-8,
                 weight_decay: float = 0.0):
        self.m = manager
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        pshards, _ = manager.shards()
        self.exp_avgs = [torch.zeros_like(p, dtype=torch.float32)
                         for p in pshards]
        self.exp_avg_sqs = [torch.zeros_like(p, dtype=torch.float32)
                            for p in pshards]

    def moment_slices(self):
        """Per-parameter views of the sharded moments (see
        ZeroOptimizer.moment_slices): (param, lo, hi, m_view, v_view)."""
        out = []
        for bi, blk in enumerate(self.m.blocks):
            s_lo = blk.rank_idx * blk.shard_n
            s_hi = s_lo + blk.shard_n
            for p, off, n in zip(blk.params, blk.offsets, blk.numels):
                a, c = max(off, s_lo), min(off + n, s_hi)
                if a < c:
                    out.append((p, a - off, c - off,
                                self.exp_avgs[bi][a - s_lo:c - s_lo],
                                self.exp_avg_sqs[bi][a - s_lo:c - s_lo]))
        return out

    @torch.no_grad()
    def step(self, grads=None, grad_scale: float = 1.0):
        self.step_count += 1
        from ..ops import fp8 as _fp8
        _fp8.bump_epoch()  # invalidate fp8 quantized-weight caches
        pshards, gshards = self.m.shards()
        ops.fused_adamw(pshards, gshards, self.exp_avgs, self.exp_avg_sqs,
                        self.step_count, self.lr, self.beta1, self.beta2,
                        self.eps, self.weight_decay, grad_scale)

    def zero_grad(self):
        self.m.zero_grads()

    def state_dict(self):
        return {"step": self.step_count, "exp_avgs": self.exp_avgs,
                "exp_avg_sqs": self.exp_avg_sqs}

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        for d, s in zip(self.exp_avgs, sd["exp_avgs"]):
            d.copy_(s)
        for d, s in zip(self.exp_avg_sqs, sd["exp_avg_sqs"]):
            d.copy_(s)
