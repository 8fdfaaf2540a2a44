"""GPT model family (decoder-only transformer), MI355X-first.

Capability analog of the reference's Flax GPT (``alpa/model/gpt_model.py:19``
builds a BERT-style decoder + LM head) and the benchmark spec ladder
(``benchmark/alpa/suite_manual_gpt.py:16-28``): 125M…76B at seq 1024,
vocab 51200.

Design: plain nn.Modules over the hand-written op library (fused LayerNorm,
flash attention, bias+GeLU epilogue, fused vocab cross-entropy).  Tensor
parallelism is built into the layers via a DeviceMesh axis — the
auto-sharding planner picks (dp, tp) and constructs the model accordingly.
Every matmul is hipBLASLt via torch.matmul; everything fusable is a gfx950
HIP kernel.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..mesh import DeviceMesh
from ..parallel.layers import (ColumnParallelLinear, RowParallelLinear,
                               VocabParallelEmbedding,
                               vocab_parallel_cross_entropy)


@dataclass
class GPTConfig:
    hidden_size: int = 1024
    num_layers: int = 12
    num_heads: int = 16
    seq_len: int = 1024
    vocab_size: int = 51200
    ffn_mult: int = 4
    layernorm_eps: float = 1e-5
    tie_embeddings: bool = False

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_heads

    def num_params(self) -> int:
        h, l, v = self.hidden_size, self.num_layers, self.vocab_size
        per_layer = 4 * h * h + 2 * self.ffn_mult * h * h
        return l * per_layer + v * h + self.seq_len * h + v * h


# The reference's GPT spec ladder (suite_manual_gpt.py:16-28):
# name -> (hidden, layers, heads)
GPT_SPECS = {
    "125M": (768, 12, 12),
    "350M": (1024, 24, 16),
    "760M": (1536, 24, 16),
    "1.3B": (2048, 24, 32),
    "2.6B": (2560, 32, 32),
    "6.7B": (4096, 32, 32),
    "15B": (5120, 48, 40),
    "39B": (8192, 48, 64),
    "76B": (10240, 60, 80),
}


def gpt_config(name: str, seq_len: int = 1024,
               vocab_size: int = 51200) -> GPTConfig:
    h, l, heads = GPT_SPECS[name]
    return GPTConfig(hidden_size=h, num_layers=l, num_heads=heads,
                     seq_len=seq_len, vocab_size=vocab_size)


class Attention(nn.Module):
    """Multi-head attention: fused qkv column-split by heads over the tp
    axis, flash-attention kernel, output row-split + all-reduce."""

    def __init__(self, cfg: GPTConfig, mesh: Optional[DeviceMesh], axis: int,
                 dtype, device):
        super().__init__()
        self.cfg = cfg
        tp = mesh.axis_size(axis) if mesh is not None else 1
        assert cfg.num_heads % tp == 0
        self.heads_per_rank = cfg.num_heads // tp
        self.head_dim = cfg.head_dim
        self.qkv = ColumnParallelLinear(cfg.hidden_size, 3 * cfg.hidden_size,
                                        mesh, axis, dtype=dtype, device=device)
        self.out = RowParallelLinear(cfg.hidden_size, cfg.hidden_size, mesh,
                                     axis, dtype=dtype, device=device)

    def forward(self, x):
        # packed per-head [q|k|v] projection; the strided attention kernel
        # consumes/produces these layouts directly (no permute copies)
        qkv = self.qkv(x)  # [B, S, 3*H/tp]
        o = ops.flash_attention_qkv(qkv, self.heads_per_rank, causal=True)
        return self.out(o)


class MLP(nn.Module):

    def __init__(self, cfg: GPTConfig, mesh: Optional[DeviceMesh], axis: int,
                 dtype, device):
        super().__init__()
        ffn = cfg.ffn_mult * cfg.hidden_size
        self.fc1 = ColumnParallelLinear(cfg.hidden_size, ffn, mesh, axis,
                                        gelu=True, dtype=dtype, device=device)
        self.fc2 = RowParallelLinear(ffn, cfg.hidden_size, mesh, axis,
                                     dtype=dtype, device=device)

    def forward(self, x):
        return self.fc2(self.fc1(x))


class LayerNorm(nn.Module):

    def __init__(self, hidden: int, eps: float, dtype, device):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(hidden, dtype=dtype,
                                              device=device))
        self.bias = nn.Parameter(torch.zeros(hidden, dtype=dtype,
                                             device=device))

    def forward(self, x):
        return ops.layer_norm(x, self.weight, self.bias, self.eps)


class Block(nn.Module):

    def __init__(self, cfg: GPTConfig, mesh, axis, dtype, device):
        super().__init__()
        self.ln1 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype, device)
        self.attn = Attention(cfg, mesh, axis, dtype, device)
        self.ln2 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype, device)
        self.mlp = MLP(cfg, mesh, axis, dtype, device)

    def forward(self, x):
        x = x + self.attn(self.ln1(x))
        x = x + self.mlp(self.ln2(x))
        return x


class GPTModel(nn.Module):
    """Full GPT LM. `mesh`/`axis` give the tensor-parallel axis (axis 1 of a
    (dp, tp) mesh); dp replication is handled by the trainer."""

    def __init__(self, cfg: GPTConfig, mesh: Optional[DeviceMesh] = None,
                 axis: int = 1, dtype=torch.float32, device=None):
        super().__init__()
        self.cfg = cfg
        self.mesh, self.axis = mesh, axis
        self.wte = VocabParallelEmbedding(cfg.vocab_size, cfg.hidden_size,
                                          mesh, axis, dtype=dtype,
                                          device=device)
        self.wpe = nn.Parameter(
            torch.empty(cfg.seq_len, cfg.hidden_size, dtype=dtype,
                        device=device).normal_(0, 0.02))
        self.blocks = nn.ModuleList([
            Block(cfg, mesh, axis, dtype, device)
            for _ in range(cfg.num_layers)
        ])
        self.ln_f = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                              device)
        # LM head: column-parallel over vocab; loss is computed shard-local
        # via vocab-parallel cross-entropy (no logits gather).
        self.lm_head = ColumnParallelLinear(cfg.hidden_size, cfg.vocab_size,
                                            mesh, axis, bias=False,
                                            dtype=dtype, device=device)
        if cfg.tie_embeddings:
            self.lm_head.weight = self.wte.weight

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        """ids [B, S] -> logits [B, S, V/tp] (vocab-sharded)."""
        B, S = ids.shape
        x = self.wte(ids) + self.wpe[:S]
        for blk in self.blocks:
            x = blk(x)
        x = self.ln_f(x)
        return self.lm_head(x)

    def loss(self, ids: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
        """Mean per-token LM loss. labels [B, S]."""
        logits = self.forward(ids)
        N = logits.shape[0] * logits.shape[1]
        logits = logits.reshape(N, -1)
        vocab_start = self.lm_head.mesh.axis_index(self.axis) * \
            self.lm_head.out_per_rank if self.mesh is not None else 0
        per_tok = vocab_parallel_cross_entropy(
            logits, labels.reshape(N), self.mesh, self.axis, vocab_start)
        return per_tok.mean()
