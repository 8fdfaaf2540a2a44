"""Fused AdamW optimizer with fp32 state over bf16 params.

The update itself is one multi-tensor gfx950 HIP kernel launch per step
(ops.fused_adamw) — the analog of the fused Adam inside the reference's
apply_grad HLO.  Grad scaling (1/num_microbatches,
1/dp for summed all-reduce) is folded into the kernel so no separate
grad-division pass over HBM is needed (reference folds it into the jaxpr:
``shard_parallel/compile_executable.py:272`` "grad / num_micro_batches").
"""
from __future__ import annotations

from typing import Iterable, List, Optional

import torch

from . import ops


class AdamW:
    """Minimal AdamW over an explicit param list.

    `grads` may be supplied (e.g. views into flat grad-sync buckets) or
    defaults to `p.grad`.

    Known deviation from torch.optim.AdamW: bias correction uses ONE
    global step counter.  A param that skips steps (no grad) and later
    rejoins is corrected with the global t — slightly weaker correction
    for its first real updates.  Training steps through `parallelize`
    always populate every grad via the bucket views, so the deviation
    only affects the raw-optimizer path with genuinely unused params.
    """

    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float = 1e-4,
                 betas=(0.9, 0.95), eps: float = 1e-8,
                 weight_decay: float = 0.01):
        self.params: List[torch.nn.Parameter] = [
            p for p in params if p.requires_grad
        ]
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.exp_avgs = [
            torch.zeros(p.shape, dtype=torch.float32, device=p.device)
            for p in self.params
        ]
        self.exp_avg_sqs = [
            torch.zeros(p.shape, dtype=torch.float32, device=p.device)
            for p in self.params
        ]

    @torch.no_grad()
    def step(self, grads: Optional[List[torch.Tensor]] = None,
             grad_scale: float = 1.0):
        if grads is None:
            grads = [p.grad for p in self.params]
        self.step_count += 1
        from .ops import fp8 as _fp8
        _fp8.bump_epoch()  # invalidate fp8 quantized-weight caches
        # params without a grad this step (e.g. an unused head when
        # training without the grad-sync bucket views) are skipped
        quads = [(p, g, m, v) for p, g, m, v in
                 zip(self.params, grads, self.exp_avgs, self.exp_avg_sqs)
                 if g is not None]
        if not quads:
            return
        ps, gs, ms, vs = (list(t) for t in zip(*quads))
        ops.fused_adamw(ps, gs, ms, vs, self.step_count, self.lr,
                        self.beta1, self.beta2, self.eps,
                        self.weight_decay, grad_scale)

    def step_sharded(self, grad_sync, grad_scale: float = 1.0):
        """ZeRO-2 path: installed by parallel/zero.py (sharded state over
        reduce-scattered flat grads)."""
        raise NotImplementedError(
            "ZeRO-2 requires ZeroOptimizer (alpa_amd.parallel.zero)")

    def zero_grad(self):
        for p in self.params:
            if p.grad is not None:
                p.grad.zero_()

    # ------- checkpointing -------
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
