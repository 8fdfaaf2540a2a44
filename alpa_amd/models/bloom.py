"""BLOOM decoder family: ALiBi attention, no positional embeddings.

Capability analog of the reference's ``examples/llm_serving/model/
bloom_model.py`` (Flax BLOOM with cache).  ALiBi is computed INSIDE the
gfx950 attention kernels (ops/csrc/attention.hip: per-head slope fma on
the score tile — fwd, bwd-dq and bwd-dkv), so both training and cached
decode stay on the fused path with zero bias materialization.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..mesh import DeviceMesh
from ..parallel.layers import (ColumnParallelLinear, RowParallelLinear,
                               VocabParallelEmbedding)
from .generation import GenerationMixin
from .gpt import LayerNorm
from .opt import KVCache


@dataclass
class BloomConfig:
    hidden_size: int = 1024
    num_layers: int = 24
    num_heads: int = 16
    vocab_size: int = 50304
    max_seq_len: int = 2048
    ffn_mult: int = 4
    layernorm_eps: float = 1e-5

    @property
    def head_dim(self):
        return self.hidden_size // self.num_heads


# public BLOOM ladder
BLOOM_SPECS = {
    "560M": (1024, 24, 16),
    "1.7B": (2048, 24, 16),
    "3B": (2560, 30, 32),
    "7.1B": (4096, 30, 32),
    "176B": (14336, 70, 112),
}


def bloom_config(name: str, max_seq_len: int = 2048) -> BloomConfig:
    h, l, heads = BLOOM_SPECS[name]
    return BloomConfig(hidden_size=h, num_layers=l, num_heads=heads,
                       max_seq_len=max_seq_len)


def alibi_slopes(num_heads: int) -> torch.Tensor:
    """Per-head ALiBi slopes (the BLOOM/press-et-al geometric ladder):
    for 2^k heads, slope_h = 2^(-8(h+1)/H); non-powers-of-two interleave
    the next power's odd ladder."""
    def pow2_slopes(n):
        start = 2.0 ** (-(2.0 ** -(math.log2(n) - 3)))
        return [start * (start ** i) for i in range(n)]

    if math.log2(num_heads).is_integer():
        out = pow2_slopes(num_heads)
    else:
        base = 2 ** math.floor(math.log2(num_heads))
        out = pow2_slopes(base)
        extra = pow2_slopes(2 * base)[0::2][:num_heads - base]
        out = out + extra
    return torch.tensor(out, dtype=torch.float32)


class BloomBlock(nn.Module):

    def __init__(self, cfg: BloomConfig, mesh, axis, dtype, device, idx,
                 init_seed):
        super().__init__()
        tp = mesh.axis_size(axis) if mesh is not None else 1
        self.heads_per_rank = cfg.num_heads // tp
        self.head_dim = cfg.head_dim
        self.ln1 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                             device)
        self.qkv = ColumnParallelLinear(cfg.hidden_size, 3 * cfg.hidden_size,
                                        mesh, axis, dtype=dtype,
                                        device=device, init_seed=init_seed,
                                        init_tag=f"b{idx}.qkv")
        self.out = RowParallelLinear(cfg.hidden_size, cfg.hidden_size, mesh,
                                     axis, dtype=dtype, device=device,
                                     init_seed=init_seed,
                                     init_tag=f"b{idx}.out")
        self.ln2 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                             device)
        self.fc1 = ColumnParallelLinear(cfg.hidden_size,
                                        cfg.ffn_mult * cfg.hidden_size,
                                        mesh, axis, gelu=True, dtype=dtype,
                                        device=device, init_seed=init_seed,
                                        init_tag=f"b{idx}.fc1")
        self.fc2 = RowParallelLinear(cfg.ffn_mult * cfg.hidden_size,
                                     cfg.hidden_size, mesh, axis,
                                     dtype=dtype, device=device,
                                     init_seed=init_seed,
                                     init_tag=f"b{idx}.fc2")

    def _attn(self, x, cache_k, cache_v, start_pos: int, slopes):
        B, S, _ = x.shape
        h, d = self.heads_per_rank, self.head_dim
        qkv = self.qkv(x).view(B, S, h, 3, d)
        q = qkv[:, :, :, 0].permute(0, 2, 1, 3)
        k = qkv[:, :, :, 1].permute(0, 2, 1, 3)
        v = qkv[:, :, :, 2].permute(0, 2, 1, 3)
        cache_k[:, :, start_pos:start_pos + S] = k
        cache_v[:, :, start_pos:start_pos + S] = v
        total = start_pos + S
        kc = cache_k[:, :, :total]
        vc = cache_v[:, :, :total]
        o = ops.flash_attention(q.contiguous(), kc, vc,
                                causal=(S == total and S > 1),
                                alibi_slopes=slopes)
        return self.out(o.permute(0, 2, 1, 3).reshape(B, S, h * d))

    def forward(self, x, cache_k, cache_v, start_pos: int, slopes):
        x = x + self._attn(self.ln1(x), cache_k, cache_v, start_pos,
                           slopes)
        return x + self.fc2(self.fc1(self.ln2(x)))


class BloomModel(nn.Module, GenerationMixin):
    """TP-sharded BLOOM decoder with KV-cache generation.  Each rank's
    slopes vector covers ITS heads (the TP shard offsets into the global
    slope ladder)."""

    def __init__(self, cfg: BloomConfig, mesh: Optional[DeviceMesh] = None,
                 axis: int = 1, dtype=torch.float32, device=None,
                 init_seed: int = 0):
        super().__init__()
        self.cfg = cfg
        assert cfg.head_dim <= 128, (
            f"head_dim {cfg.head_dim} > 128: the gfx950 attention kernel "
            "tiles head_dim in registers up to 128 (split-D is a planned "
            "extension); pick a config with head_dim <= 128")
        self.mesh, self.axis = mesh, axis
        tp = mesh.axis_size(axis) if mesh is not None else 1
        self.heads_per_rank = cfg.num_heads // tp
        idx = mesh.axis_index(axis) if (mesh is not None and
                                        mesh.is_member) else 0
        idx = max(idx, 0)
        slopes = alibi_slopes(cfg.num_heads)[
            idx * self.heads_per_rank:(idx + 1) * self.heads_per_rank]
        self.register_buffer("slopes", slopes.to(device=device),
                             persistent=False)
        self.wte = VocabParallelEmbedding(cfg.vocab_size, cfg.hidden_size,
                                          mesh, axis, dtype=dtype,
                                          device=device,
                                          init_seed=init_seed,
                                          init_tag="wte")
        # BLOOM normalizes the embeddings before the first block
        self.ln_emb = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                                device)
        self.blocks = nn.ModuleList([
            BloomBlock(cfg, mesh, axis, dtype, device, i, init_seed)
            for i in range(cfg.num_layers)
        ])
        self.ln_f = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                              device)
        self.lm_head = ColumnParallelLinear(cfg.hidden_size, cfg.vocab_size,
                                            mesh, axis, bias=False,
                                            dtype=dtype, device=device,
                                            init_seed=init_seed,
                                            init_tag="lm_head")
        self.lm_head._fp8_exclude = True  # logits GEMM stays bf16
        self.dtype = dtype
        self.device_ = device

    def new_cache(self, batch: int, max_len: Optional[int] = None
                  ) -> KVCache:
        return KVCache(self.cfg, self.cfg.num_layers, batch,
                       self.heads_per_rank, self.dtype, self.device_,
                       max_len=max_len)

    @torch.no_grad()
    def forward_prefill(self, ids: torch.Tensor, lens: torch.Tensor,
                        cache: KVCache) -> torch.Tensor:
        """Batched variable-length prefill (see OPTModel.forward_prefill);
        ALiBi biases depend only on relative positions, so right-padded
        causal prefill stays exact for rows < len."""
        B, S = ids.shape
        x = self.ln_emb(self.wte(ids))
        for i, blk in enumerate(self.blocks):
            x = blk(x, cache.k[i], cache.v[i], 0, self.slopes)
        cache.length = S
        idx = (lens.to(ids.device) - 1).clamp(min=0)
        x_last = x[torch.arange(B, device=ids.device), idx].unsqueeze(1)
        return self.lm_head(self.ln_f(x_last))[:, 0]

    def forward_step(self, ids: torch.Tensor, cache: KVCache
                     ) -> torch.Tensor:
        B, S = ids.shape
        pos = cache.length
        assert pos == 0 or S == 1, "chunked decode with history unsupported"
        x = self.ln_emb(self.wte(ids))
        for i, blk in enumerate(self.blocks):
            x = blk(x, cache.k[i], cache.v[i], pos, self.slopes)
        cache.length += S
        return self.lm_head(self.ln_f(x[:, -1:]))[:, 0]
