"""GShard-style MoE transformer (reference ``alpa/model/moe.py``:
top2_gating:85, FlaxPositionWiseMoELayer:144, FlaxMoELayerCollection:231;
benchmarked in suite_auto_moe.py — BASELINE config 4).

Every other block's FFN is a top-2 gated expert layer; experts are
sharded over the mesh's dp axis (expert parallelism), attention/MLP may
still be tensor-parallel over axis 1.  Dispatch/combine are RCCL
all-to-all over xGMI (parallel/expert.py).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from ..mesh import DeviceMesh
from ..parallel.expert import ExpertParallelMLP
from .gpt import (Attention, GPTConfig, LayerNorm, _wpe_init,
                  ColumnParallelLinear, VocabParallelEmbedding,
                  vocab_parallel_cross_entropy)


@dataclass
class MoEConfig(GPTConfig):
    num_experts: int = 8
    moe_every: int = 2          # every k-th block uses the MoE FFN
    capacity_factor: float = 2.0
    aux_loss_weight: float = 1e-2
    #: mesh axis carrying expert parallelism (dp axis by default)
    ep_axis: int = 0


class MoEBlock(nn.Module):

    def __init__(self, cfg: MoEConfig, mesh, axis, dtype, device,
                 layer_idx: int, init_seed: int):
        super().__init__()
        self.ln1 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                             device)
        self.attn = Attention(cfg, mesh, axis, dtype, device, layer_idx,
                              init_seed)
        self.ln2 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                             device)
        self.moe = ExpertParallelMLP(
            cfg.hidden_size, cfg.ffn_mult * cfg.hidden_size,
            cfg.num_experts, mesh, cfg.ep_axis,
            capacity_factor=cfg.capacity_factor, dtype=dtype, device=device,
            layer_idx=layer_idx, init_seed=init_seed)

    def forward(self, x):
        x = x + self.attn(self.ln1(x))
        x = x + self.moe(self.ln2(x))
        return x


class MoEGPTModel(nn.Module):
    """Decoder LM with interleaved dense/MoE blocks."""

    def __init__(self, cfg: MoEConfig, mesh: Optional[DeviceMesh] = None,
                 axis: int = 1, dtype=torch.float32, device=None,
                 init_seed: int = 0):
        super().__init__()
        from .gpt import Block
        self.cfg = cfg
        self.mesh, self.axis = mesh, axis
        self.wte = VocabParallelEmbedding(cfg.vocab_size, cfg.hidden_size,
                                          mesh, axis, dtype=dtype,
                                          device=device, init_seed=init_seed,
                                          init_tag="wte")
        self.wpe = nn.Parameter(_wpe_init(cfg, dtype, device, init_seed))
        blocks = []
        for i in range(cfg.num_layers):
            if (i + 1) % cfg.moe_every == 0:
                blocks.append(MoEBlock(cfg, mesh, axis, dtype, device, i,
                                       init_seed))
            else:
                blocks.append(Block(cfg, mesh, axis, dtype, device, i,
                                    init_seed))
        self.blocks = nn.ModuleList(blocks)
        self.ln_f = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                              device)
        self.lm_head = ColumnParallelLinear(cfg.hidden_size, cfg.vocab_size,
                                            mesh, axis, bias=False,
                                            dtype=dtype, device=device,
                                            init_seed=init_seed,
                                            init_tag="lm_head")

    def forward(self, ids):
        B, S = ids.shape
        x = self.wte(ids) + self.wpe[:S]
        for blk in self.blocks:
            x = blk(x)
        return self.lm_head(self.ln_f(x))

    def loss(self, ids, labels):
        logits = self.forward(ids)
        N = logits.shape[0] * logits.shape[1]
        logits = logits.reshape(N, -1)
        vocab_start = self.lm_head.mesh.axis_index(self.axis) * \
            self.lm_head.out_per_rank if self.mesh is not None else 0
        per_tok = vocab_parallel_cross_entropy(
            logits, labels.reshape(N), self.mesh, self.axis, vocab_start)
        loss = per_tok.mean()
        # auxiliary load-balancing loss from every MoE layer
        aux = None
        for blk in self.blocks:
            if isinstance(blk, MoEBlock) and blk.moe.last_aux_loss is not None:
                aux = blk.moe.last_aux_loss if aux is None \
                    else aux + blk.moe.last_aux_loss
        if aux is not None:
            loss = loss + self.cfg.aux_loss_weight * aux.to(loss.dtype)
        return loss
