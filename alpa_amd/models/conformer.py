"""Conformer encoder block/model (reference ``alpa/model/conformer.py``):
feed-forward half-step sandwich around self-attention + depthwise
convolution module — the speech-model entry of the zoo.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .gpt import LayerNorm


@dataclass
class ConformerConfig:
    hidden_size: int = 256
    num_layers: int = 4
    num_heads: int = 4
    ffn_mult: int = 4
    conv_kernel: int = 15
    layernorm_eps: float = 1e-5


class FeedForwardModule(nn.Module):

    def __init__(self, cfg, dtype, device):
        super().__init__()
        h, f = cfg.hidden_size, cfg.ffn_mult * cfg.hidden_size
        self.ln = LayerNorm(h, cfg.layernorm_eps, dtype, device)
        self.w1 = nn.Linear(h, f)
        self.w2 = nn.Linear(f, h)
        self.to(dtype=dtype)
        if device is not None:
            self.to(device)

    def forward(self, x):
        return self.w2(F.silu(self.w1(self.ln(x))))


class ConvModule(nn.Module):
    """Pointwise-GLU -> depthwise conv -> norm -> swish -> pointwise."""

    def __init__(self, cfg, dtype, device):
        super().__init__()
        h = cfg.hidden_size
        self.ln = LayerNorm(h, cfg.layernorm_eps, dtype, device)
        self.pw1 = nn.Conv1d(h, 2 * h, 1)
        self.dw = nn.Conv1d(h, h, cfg.conv_kernel,
                            padding=cfg.conv_kernel // 2, groups=h)
        self.bn = nn.GroupNorm(1, h)
        self.pw2 = nn.Conv1d(h, h, 1)
        self.to(dtype=dtype)
        if device is not None:
            self.to(device)

    def forward(self, x):
        y = self.ln(x).transpose(1, 2)      # [B, H, S]
        y = F.glu(self.pw1(y), dim=1)
        y = self.pw2(F.silu(self.bn(self.dw(y))))
        return y.transpose(1, 2)


class ConformerBlock(nn.Module):

    def __init__(self, cfg, dtype, device):
        super().__init__()
        h = cfg.hidden_size
        self.ff1 = FeedForwardModule(cfg, dtype, device)
        self.ln_attn = LayerNorm(h, cfg.layernorm_eps, dtype, device)
        self.qkv = nn.Linear(h, 3 * h)
        self.attn_out = nn.Linear(h, h)
        self.heads = cfg.num_heads
        self.conv = ConvModule(cfg, dtype, device)
        self.ff2 = FeedForwardModule(cfg, dtype, device)
        self.ln_out = LayerNorm(h, cfg.layernorm_eps, dtype, device)
        self.to(dtype=dtype)
        if device is not None:
            self.to(device)

    def forward(self, x):
        x = x + 0.5 * self.ff1(x)
        a = self.qkv(self.ln_attn(x))
        a = ops.flash_attention_qkv(a, self.heads, causal=False)
        x = x + self.attn_out(a)
        x = x + self.conv(x)
        x = x + 0.5 * self.ff2(x)
        return self.ln_out(x)


class ConformerEncoder(nn.Module):

    def __init__(self, cfg: ConformerConfig, input_dim: int = 80,
                 dtype=torch.float32, device=None):
        super().__init__()
        self.proj = nn.Linear(input_dim, cfg.hidden_size)
        self.blocks = nn.ModuleList([
            ConformerBlock(cfg, dtype, device)
            for _ in range(cfg.num_layers)
        ])
        self.to(dtype=dtype)
        if device is not None:
            self.to(device)

    def forward(self, feats):
        """feats [B, S, input_dim] -> [B, S, hidden]."""
        x = self.proj(feats)
        for blk in self.blocks:
            x = blk(x)
        return x
