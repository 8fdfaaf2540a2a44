"""CodeGen decoder family (GPT-J architecture) with rotary positions.

Capability analog of the reference's ``examples/llm_serving/model/
codegen_model.py`` (Flax CodeGen with cache).  GPT-J-style blocks:
ONE LayerNorm feeding attention AND the MLP in parallel
(x + attn(ln(x)) + mlp(ln(x))), rotary embedding applied to the first
``rotary_dim`` dims of every head (interleaved/"rotate_every_two"
convention).  Rotary is an elementwise pre-transform of q/k, so the
unmodified gfx950 flash-attention kernel serves both prefill and cached
decode; generation (greedy + beam) comes from GenerationMixin.
"""
from __future__ import annotations

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
class CodeGenConfig:
    hidden_size: int = 1024
    num_layers: int = 20
    num_heads: int = 16
    vocab_size: int = 51200
    max_seq_len: int = 2048
    rotary_dim: int = 32
    ffn_mult: int = 4
    layernorm_eps: float = 1e-5

    @property
    def head_dim(self):
        return self.hidden_size // self.num_heads


# public CodeGen ladder (350M/2B/6B/16B)
CODEGEN_SPECS = {
    "350M": (1024, 20, 16),
    "2B": (2560, 32, 32),
    "6B": (4096, 33, 16),
    "16B": (6144, 34, 24),
}


def codegen_config(name: str, max_seq_len: int = 2048) -> CodeGenConfig:
    h, l, heads = CODEGEN_SPECS[name]
    return CodeGenConfig(hidden_size=h, num_layers=l, num_heads=heads,
                         max_seq_len=max_seq_len)


def rotary_tables(rotary_dim: int, max_len: int, dtype, device):
    """(sin, cos) tables [max_len, rotary_dim//2] — GPT-J inv_freq."""
    inv = 1.0 / (10000.0 ** (torch.arange(0, rotary_dim, 2,
                                          dtype=torch.float32) / rotary_dim))
    t = torch.arange(max_len, dtype=torch.float32)
    freqs = torch.outer(t, inv)
    return (freqs.sin().to(dtype=dtype, device=device),
            freqs.cos().to(dtype=dtype, device=device))


def apply_rotary(x: torch.Tensor, sin: torch.Tensor, cos: torch.Tensor,
                 pos: int, rotary_dim: int) -> torch.Tensor:
    """x [B, h, S, d]: rotate the first rotary_dim dims, interleaved pairs
    (GPT-J "rotate_every_two"); positions pos..pos+S-1."""
    B, H, S, D = x.shape
    r = x[..., :rotary_dim].view(B, H, S, rotary_dim // 2, 2)
    x1, x2 = r[..., 0], r[..., 1]
    s = sin[pos:pos + S].view(1, 1, S, -1).to(x.dtype)
    c = cos[pos:pos + S].view(1, 1, S, -1).to(x.dtype)
    rot = torch.stack([x1 * c - x2 * s, x1 * s + x2 * c], dim=-1)
    return torch.cat([rot.flatten(-2), x[..., rotary_dim:]], dim=-1)


class CodeGenBlock(nn.Module):
    """Parallel-residual block: x + attn(ln x) + mlp(ln x) (GPT-J)."""

    def __init__(self, cfg: CodeGenConfig, mesh, axis, dtype, device, idx,
                 init_seed):
        super().__init__()
        tp = mesh.axis_size(axis) if mesh is not None else 1
        self.heads_per_rank = cfg.num_heads // tp
        self.head_dim = cfg.head_dim
        self.rotary_dim = cfg.rotary_dim
        self.ln = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                            device)
        self.qkv = ColumnParallelLinear(cfg.hidden_size, 3 * cfg.hidden_size,
                                        mesh, axis, dtype=dtype,
                                        device=device, init_seed=init_seed,
                                        init_tag=f"b{idx}.qkv")
        self.out = RowParallelLinear(cfg.hidden_size, cfg.hidden_size, mesh,
                                     axis, dtype=dtype, device=device,
                                     init_seed=init_seed,
                                     init_tag=f"b{idx}.out")
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

    def _attn(self, y, cache_k, cache_v, start_pos: int, sin, cos):
        B, S, _ = y.shape
        h, d = self.heads_per_rank, self.head_dim
        qkv = self.qkv(y).view(B, S, h, 3, d)
        q = qkv[:, :, :, 0].permute(0, 2, 1, 3)
        k = qkv[:, :, :, 1].permute(0, 2, 1, 3)
        v = qkv[:, :, :, 2].permute(0, 2, 1, 3)
        q = apply_rotary(q, sin, cos, start_pos, self.rotary_dim)
        k = apply_rotary(k, sin, cos, start_pos, self.rotary_dim)
        cache_k[:, :, start_pos:start_pos + S] = k
        cache_v[:, :, start_pos:start_pos + S] = v
        total = start_pos + S
        kc = cache_k[:, :, :total]
        vc = cache_v[:, :, :total]
        o = ops.flash_attention(q.contiguous(), kc, vc,
                                causal=(S == total and S > 1))
        return self.out(o.permute(0, 2, 1, 3).reshape(B, S, h * d))

    def forward(self, x, cache_k, cache_v, start_pos: int, sin, cos):
        y = self.ln(x)
        return x + self._attn(y, cache_k, cache_v, start_pos, sin, cos) + \
            self.fc2(self.fc1(y))


class CodeGenModel(nn.Module, GenerationMixin):
    """TP-sharded CodeGen decoder with KV-cache generation."""

    def __init__(self, cfg: CodeGenConfig, mesh: Optional[DeviceMesh] = None,
                 axis: int = 1, dtype=torch.float32, device=None,
                 init_seed: int = 0):
        super().__init__()
        self.cfg = cfg
        assert cfg.head_dim <= 256, (
            f"head_dim {cfg.head_dim} > 256 unsupported; dims in "
            "(128, 256] use the blocked hipBLASLt attention path")
        self.mesh, self.axis = mesh, axis
        tp = mesh.axis_size(axis) if mesh is not None else 1
        self.heads_per_rank = cfg.num_heads // tp
        self.wte = VocabParallelEmbedding(cfg.vocab_size, cfg.hidden_size,
                                          mesh, axis, dtype=dtype,
                                          device=device,
                                          init_seed=init_seed,
                                          init_tag="wte")
        self.blocks = nn.ModuleList([
            CodeGenBlock(cfg, mesh, axis, dtype, device, i, init_seed)
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
        sin, cos = rotary_tables(cfg.rotary_dim, cfg.max_seq_len, dtype,
                                 device)
        self.register_buffer("rot_sin", sin, persistent=False)
        self.register_buffer("rot_cos", cos, persistent=False)
        self.dtype = dtype
        self.device_ = device

    def new_cache(self, batch: int) -> KVCache:
        return KVCache(self.cfg, self.cfg.num_layers, batch,
                       self.heads_per_rank, self.dtype, self.device_)

    def forward_step(self, ids: torch.Tensor, cache: KVCache
                     ) -> torch.Tensor:
        B, S = ids.shape
        pos = cache.length
        assert pos == 0 or S == 1, "chunked decode with history unsupported"
        x = self.wte(ids)
        for i, blk in enumerate(self.blocks):
            x = blk(x, cache.k[i], cache.v[i], pos, self.rot_sin,
                    self.rot_cos)
        cache.length += S
        return self.lm_head(self.ln_f(x[:, -1:]))[:, 0]
