"""ZeRO-2/3 over the flat grad buckets (reference: Zero2Parallel/Zero3Parallel,
parallel_method.py:130,146; mechanism = prefer_reduce_scatter +
optimizer-state partitioning, auto_sharding.py:69,290-295).

ZeRO-2 here:
- grads are reduce-scattered per bucket (GradSynchronizer reduce_scatter
  mode) so each dp rank owns 1/dp of every grad bucket;
- parameters are flattened into bucket-matched flat buffers (each param's
  ``.data`` becomes a view), the fused AdamW kernel updates only this
  rank's shard (fp32 moments exist only for the shard — the memory win);
- updated params are all-gathered per bucket, overlappable on the comm
  stream.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from .. import ops
from ..mesh import is_distributed
from .grad_sync import GradSynchronizer


class ZeroOptimizer:
    """Sharded AdamW over GradSynchronizer's reduce-scatter buckets."""

    def __init__(self, grad_sync: GradSynchronizer, lr: float = 1e-4,
                 betas=(0.9, 0.95), eps: float = 1e-8,
                 weight_decay: float = 0.0):
        self.gs = grad_sync
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.dp = grad_sync.dp
        mesh, axis = grad_sync.mesh, grad_sync.axis
        self.rank_idx = mesh.axis_index(axis) if (mesh is not None and
                                                  self.dp > 1) else 0

        # flatten params into bucket-image flat buffers
        self.flat_params: List[torch.Tensor] = []
        self.param_shards: List[torch.Tensor] = []
        self.grad_shards: List[torch.Tensor] = []
        self.exp_avgs: List[torch.Tensor] = []
        self.exp_avg_sqs: List[torch.Tensor] = []
        for b in self.gs.buckets:
            total = b.flat.numel()
            flatp = torch.zeros(total, dtype=b.params[0].dtype,
                                device=b.params[0].device)
            off = 0
            with torch.no_grad():
                for p in b.params:
                    n = p.numel()
                    flatp[off:off + n].copy_(p.data.reshape(-1))
                    p.data = flatp[off:off + n].view(p.shape)
                    off += n
            self.flat_params.append(flatp)
            if self.gs.reduce_scatter:
                shard_n = total // self.dp
                pshard = flatp[self.rank_idx * shard_n:
                               (self.rank_idx + 1) * shard_n]
                gshard = b.shard
            else:  # dp == 1 fallback: "shard" is the whole bucket
                pshard, gshard = flatp, b.flat
            self.param_shards.append(pshard)
            self.grad_shards.append(gshard)
            self.exp_avgs.append(
                torch.zeros_like(pshard, dtype=torch.float32))
            self.exp_avg_sqs.append(
                torch.zeros_like(pshard, dtype=torch.float32))

    @property
    def params(self):
        return self.gs.params

    def moment_slices(self):
        """Per-parameter views of the sharded Adam moments, for
        checkpointing in the reference's reshard-on-load format
        (serialization.py): yields (param, lo, hi, m_view, v_view) where
        [lo, hi) is this rank's owned range within the param's OWN flat
        index space and the views alias exp_avgs/exp_avg_sqs."""
        out = []
        for bi, b in enumerate(self.gs.buckets):
            total = b.flat.numel()
            if self.gs.reduce_scatter:
                shard_n = total // self.dp
                s_lo = self.rank_idx * shard_n
                s_hi = s_lo + shard_n
            else:
                s_lo, s_hi = 0, total
            off = 0
            for p in b.params:
                n = p.numel()
                a, c = max(off, s_lo), min(off + n, s_hi)
                if a < c:
                    out.append((p, a - off, c - off,
                                self.exp_avgs[bi][a - s_lo:c - s_lo],
                                self.exp_avg_sqs[bi][a - s_lo:c - s_lo]))
                off += n
        return out

    @torch.no_grad()
    def step(self, grads=None, grad_scale: float = 1.0):
        self.step_count += 1
        from ..ops import fp8 as _fp8
        _fp8.bump_epoch()  # invalidate fp8 quantized-weight caches
        ops.fused_adamw(self.param_shards, self.grad_shards, self.exp_avgs,
                        self.exp_avg_sqs, self.step_count, self.lr,
                        self.beta1, self.beta2, self.eps, self.weight_decay,
                        grad_scale)
        # all-gather updated params per bucket
        if self.gs.reduce_scatter and is_distributed():
            group = self.gs.mesh.axis_group(self.gs.axis)
            for flatp, pshard in zip(self.flat_params, self.param_shards):
                dist.all_gather_into_tensor(flatp, pshard.contiguous(),
                                            group=group)

    def zero_grad(self):
        self.gs.zero_grads()

    def state_dict(self):
        return {
            "step": self.step_count,
            "exp_avgs": self.exp_avgs,
            "exp_avg_sqs": self.exp_avg_sqs,
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        for dst, src in zip(self.exp_avgs, sd["exp_avgs"]):
            dst.copy_(src)
        for dst, src in zip(self.exp_avg_sqs, sd["exp_avg_sqs"]):
            dst.copy_(src)
