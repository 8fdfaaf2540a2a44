"""Expert parallelism: top-k gated MoE dispatch/combine with RCCL
all-to-all over xGMI.

Capability analog of the reference's GShard MoE (``alpa/model/moe.py:85``
top2_gating, ``:144`` FlaxPositionWiseMoELayer whose dispatch/combine
einsums let the auto-sharding ILP pick expert-dim sharding -> all-to-all,
SURVEY.md §2.2 EP row).  Here the dispatch is explicit: tokens are routed
to capacity-limited expert slots, exchanged with one all-to-all per
direction (the xGMI crossbar gives every GPU 7 direct links — all-to-all
is the natural collective), processed by the local experts, exchanged
back and combined with the gate weights.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn

from ..mesh import DeviceMesh, is_distributed
from .layers import tag_seed


class _AllToAll(torch.autograd.Function):
    """Autograd-aware all-to-all (backward = all-to-all of grads)."""

    @staticmethod
    def forward(ctx, x, mesh: DeviceMesh, axis: int):
        ctx.mesh, ctx.axis = mesh, axis
        if mesh is None or mesh.axis_size(axis) == 1 or not is_distributed():
            return x.clone()
        # NB: the output buffer must be truly contiguous — empty_like on a
        # non-contiguous input (e.g. a permute's grad) inherits its strides
        # and the collective would scatter bytes into the wrong layout
        xin = x.contiguous()
        out = torch.empty_like(xin)
        dist.all_to_all_single(out, xin, group=mesh.axis_group(axis))
        return out

    @staticmethod
    def backward(ctx, g):
        mesh, axis = ctx.mesh, ctx.axis
        if mesh is None or mesh.axis_size(axis) == 1 or not is_distributed():
            return g, None, None
        gin = g.contiguous()
        out = torch.empty_like(gin)
        dist.all_to_all_single(out, gin, group=mesh.axis_group(axis))
        return out, None, None


def all_to_all(x, mesh, axis):
    return _AllToAll.apply(x, mesh, axis)


def top2_gating(logits: torch.Tensor, capacity: int
                ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor,
                           torch.Tensor]:
    """Top-2 gating with capacity (reference top2_gating, moe.py:85).

    logits [N, E] -> (combine1 [N], idx1 [N], combine2 [N], idx2 [N]).
    Slot assignment/drops are applied by the dispatcher.
    """
    gates = torch.softmax(logits.float(), dim=-1)
    top2 = gates.topk(2, dim=-1)
    w1, w2 = top2.values[:, 0], top2.values[:, 1]
    i1, i2 = top2.indices[:, 0], top2.indices[:, 1]
    denom = (w1 + w2).clamp_min(1e-9)
    return w1 / denom, i1, w2 / denom, i2


class ExpertParallelMLP(nn.Module):
    """Gated MoE FFN with experts sharded over a mesh axis.

    E experts total, E/ep per rank; dispatch/return are all-to-all over
    the ep axis.  Deterministic capacity assignment (slot order = token
    order) so EP == serial bit-for-bit up to collective reduction order.
    """

    def __init__(self, hidden: int, ffn: int, num_experts: int,
                 mesh: Optional[DeviceMesh] = None, axis: int = 1,
                 capacity_factor: float = 2.0, dtype=torch.float32,
                 device=None, layer_idx: int = 0, init_seed: int = 0):
        super().__init__()
        self.mesh, self.axis = mesh, axis
        self.ep = mesh.axis_size(axis) if mesh is not None else 1
        assert num_experts >= 2, "top-2 gating needs >= 2 experts"
        assert num_experts % self.ep == 0
        self.E = num_experts
        self.e_local = num_experts // self.ep
        self.hidden, self.ffn = hidden, ffn
        self.capacity_factor = capacity_factor
        ep_idx = mesh.axis_index(axis) if (mesh is not None and
                                           mesh.is_member) else 0
        self.e_start = max(ep_idx, 0) * self.e_local

        def init_w(shape, tag, std):
            gen_dev = device if (device is not None and
                                 torch.device(device).type == "cuda") \
                else "cpu"
            g = torch.Generator(device=gen_dev)
            g.manual_seed(tag_seed(init_seed, tag))
            w = torch.empty(shape, dtype=torch.float32, device=gen_dev)
            w.normal_(0, std, generator=g)
            return w.to(dtype=dtype, device=device)

        # gate (replicated) + local expert weights [e_local, ...]
        self.wg = nn.Parameter(init_w((hidden, num_experts),
                                      f"moe{layer_idx}.wg", 0.02))
        w1 = torch.stack([
            init_w((hidden, ffn), f"moe{layer_idx}.e{self.e_start + j}.w1",
                   1.0 / math.sqrt(hidden)) for j in range(self.e_local)])
        w2 = torch.stack([
            init_w((ffn, hidden), f"moe{layer_idx}.e{self.e_start + j}.w2",
                   1.0 / math.sqrt(ffn)) for j in range(self.e_local)])
        self.w1 = nn.Parameter(w1)
        self.w2 = nn.Parameter(w2)
        # expert weights are EP-local: token routing already delivers every
        # relevant token, so their grads are complete without the dp
        # all-reduce (GradSynchronizer honors this mark)
        self.w1._expert_parallel = True
        self.w2._expert_parallel = True
        self.last_aux_loss: Optional[torch.Tensor] = None

    def _assign_slots(self, idx: torch.Tensor, C: int,
                      base_counts: torch.Tensor):
        """Deterministic GShard-style slot assignment: token order =
        cumsum position within its expert (reference moe.py:85)."""
        mask = torch.nn.functional.one_hot(idx, self.E)  # [N, E]
        loc = torch.cumsum(mask, dim=0) - 1 + base_counts  # [N, E]
        pos = (loc * mask).sum(-1)                         # [N]
        keep = pos < C
        new_counts = base_counts + mask.sum(0, keepdim=True)
        return pos, keep, new_counts.clamp(max=C)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x [B, S, H] -> [B, S, H]."""
        B, S, H = x.shape
        N = B * S
        xf = x.reshape(N, H)
        # capacity per expert (equal all-to-all blocks)
        C = max(1, int(self.capacity_factor * N * 2 / self.E))

        logits = xf @ self.wg.to(xf.dtype)
        w1g, i1, w2g, i2 = top2_gating(logits, C)

        # aux load-balance loss (GShard): E * sum(mean_prob * frac_top1)
        gates = torch.softmax(logits.float(), dim=-1)
        me = gates.mean(dim=0)
        ce = torch.nn.functional.one_hot(i1, self.E).float().mean(dim=0)
        self.last_aux_loss = self.E * (me * ce).sum()

        zeros = torch.zeros(1, self.E, dtype=torch.long, device=x.device)
        pos1, keep1, counts = self._assign_slots(i1, C, zeros)
        pos2, keep2, _ = self._assign_slots(i2, C, counts)

        t_all = torch.arange(N, device=x.device)
        t_idx = torch.cat([t_all[keep1], t_all[keep2]])
        flat_slot = torch.cat([(i1 * C + pos1)[keep1],
                               (i2 * C + pos2)[keep2]])
        gate_w = torch.cat([w1g[keep1], w2g[keep2]])

        dispatched = xf.new_zeros(self.E * C, H)
        dispatched.index_copy_(0, flat_slot, xf.index_select(0, t_idx))

        # all-to-all: [E, C, H] blocks grouped by destination rank
        routed = all_to_all(dispatched.view(self.E, C, H), self.mesh,
                            self.axis)
        # routed block r holds rank r's tokens for OUR experts:
        # [ep, e_local, C, H] -> [e_local, ep*C, H]
        routed = routed.view(self.ep, self.e_local, C, H) \
            .transpose(0, 1).reshape(self.e_local, self.ep * C, H)

        # Per-expert 2-D GEMMs instead of one batched bmm: torch.bmm's
        # BACKWARD hits hipErrorIllegalAddress in hipBLASLt at MoE-scale
        # bf16 shapes on this stack (repro: bmm([8,8192,1024],
        # [8,1024,4096]).backward() on random tensors, ROCm 7.2 /
        # torch 2.10) — the plain mm path is unaffected and equally
        # fast at these sizes
        w1 = self.w1.to(routed.dtype)
        w2 = self.w2.to(routed.dtype)
        h = torch.stack([
            torch.nn.functional.gelu(routed[j] @ w1[j],
                                     approximate="tanh") @ w2[j]
            for j in range(self.e_local)
        ])

        # return all-to-all: back to the source ranks' [E, C, H] layout
        h = h.view(self.e_local, self.ep, C, H).transpose(0, 1) \
            .reshape(self.E, C, H)
        returned = all_to_all(h, self.mesh, self.axis)

        out = xf.new_zeros(N, H)
        out.index_add_(0, t_idx,
                       returned.view(self.E * C, H)
                       .index_select(0, flat_slot) *
                       gate_w.to(out.dtype).unsqueeze(-1))
        return out.reshape(B, S, H)
