"""BERT encoder family (reference ``alpa/model/bert_model.py`` — full BERT
incl. FlaxBertLayer:320; the reference's GPT is literally its BERT stack
with a causal flag, so the module structure is shared here too).

Bidirectional attention (causal=False flash kernel), token-type + learned
position embeddings, MLM head over the vocab-parallel embedding, pooler.
TP over the mesh axis like GPT.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..mesh import DeviceMesh
from ..parallel.layers import (ColumnParallelLinear, RowParallelLinear,
                               VocabParallelEmbedding, tag_seed,
                               vocab_parallel_cross_entropy)
from .gpt import LayerNorm, MLP


@dataclass
class BertConfig:
    hidden_size: int = 768
    num_layers: int = 12
    num_heads: int = 12
    seq_len: int = 512
    vocab_size: int = 30522
    type_vocab_size: int = 2
    ffn_mult: int = 4
    layernorm_eps: float = 1e-12


class BertAttention(nn.Module):

    def __init__(self, cfg: BertConfig, mesh, axis, dtype, device,
                 layer_idx, init_seed):
        super().__init__()
        tp = mesh.axis_size(axis) if mesh is not None else 1
        assert cfg.num_heads % tp == 0
        self.heads_per_rank = cfg.num_heads // tp
        self.qkv = ColumnParallelLinear(cfg.hidden_size, 3 * cfg.hidden_size,
                                        mesh, axis, dtype=dtype,
                                        device=device, init_seed=init_seed,
                                        init_tag=f"b{layer_idx}.qkv")
        self.out = RowParallelLinear(cfg.hidden_size, cfg.hidden_size, mesh,
                                     axis, dtype=dtype, device=device,
                                     init_seed=init_seed,
                                     init_tag=f"b{layer_idx}.out")

    def forward(self, x):
        qkv = self.qkv(x)
        o = ops.flash_attention_qkv(qkv, self.heads_per_rank, causal=False)
        return self.out(o)


class BertLayer(nn.Module):
    """Post-LN encoder block (BERT convention; reference FlaxBertLayer)."""

    def __init__(self, cfg: BertConfig, mesh, axis, dtype, device,
                 layer_idx, init_seed):
        super().__init__()
        self.attn = BertAttention(cfg, mesh, axis, dtype, device, layer_idx,
                                  init_seed)
        self.ln1 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                             device)
        self.mlp = MLP(cfg, mesh, axis, dtype, device, layer_idx, init_seed)
        self.ln2 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                             device)

    def forward(self, x):
        x = self.ln1(x + self.attn(x))
        x = self.ln2(x + self.mlp(x))
        return x


class BertModel(nn.Module):

    def __init__(self, cfg: BertConfig, mesh: Optional[DeviceMesh] = None,
                 axis: int = 1, dtype=torch.float32, device=None,
                 init_seed: int = 0):
        super().__init__()
        self.cfg = cfg
        self.mesh, self.axis = mesh, axis
        self.wte = VocabParallelEmbedding(cfg.vocab_size, cfg.hidden_size,
                                          mesh, axis, dtype=dtype,
                                          device=device, init_seed=init_seed,
                                          init_tag="wte")

        def init_p(shape, tag):
            g = torch.Generator()
            g.manual_seed(tag_seed(init_seed, tag))
            w = torch.empty(shape, dtype=torch.float32).normal_(
                0, 0.02, generator=g)
            return nn.Parameter(w.to(dtype=dtype, device=device))

        self.wpe = init_p((cfg.seq_len, cfg.hidden_size), "wpe")
        self.wtype = init_p((cfg.type_vocab_size, cfg.hidden_size), "wtype")
        self.ln_emb = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                                device)
        self.layers = nn.ModuleList([
            BertLayer(cfg, mesh, axis, dtype, device, i, init_seed)
            for i in range(cfg.num_layers)
        ])
        self.pooler = ColumnParallelLinear(cfg.hidden_size, cfg.hidden_size,
                                           None, axis, dtype=dtype,
                                           device=device,
                                           init_seed=init_seed,
                                           init_tag="pooler")
        self.lm_head = ColumnParallelLinear(cfg.hidden_size, cfg.vocab_size,
                                            mesh, axis, bias=False,
                                            dtype=dtype, device=device,
                                            init_seed=init_seed,
                                            init_tag="lm_head")

    def forward(self, ids: torch.Tensor,
                token_type_ids: Optional[torch.Tensor] = None):
        B, S = ids.shape
        x = self.wte(ids) + self.wpe[:S]
        if token_type_ids is not None:
            x = x + nn.functional.embedding(token_type_ids, self.wtype)
        x = self.ln_emb(x)
        for layer in self.layers:
            x = layer(x)
        return x

    def pooled(self, ids, token_type_ids=None):
        h = self.forward(ids, token_type_ids)
        return torch.tanh(self.pooler(h[:, 0]))

    def mlm_loss(self, ids: torch.Tensor, labels: torch.Tensor,
                 token_type_ids: Optional[torch.Tensor] = None):
        """Masked-LM loss over all positions (labels = target ids)."""
        h = self.forward(ids, token_type_ids)
        logits = self.lm_head(h)
        N = logits.shape[0] * logits.shape[1]
        logits = logits.reshape(N, -1)
        vocab_start = self.lm_head.mesh.axis_index(self.axis) * \
            self.lm_head.out_per_rank if self.mesh is not None else 0
        per_tok = vocab_parallel_cross_entropy(
            logits, labels.reshape(N), self.mesh, self.axis, vocab_start)
        return per_tok.mean()
