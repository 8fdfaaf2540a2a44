"""Vision Transformer (ViT) for image classification.

Capability analog of the reference's ``examples/ViT`` (Flax ViT fine-tune
on alpa): patch embedding + pre-LN transformer encoder (bidirectional
flash attention, causal=False) + mean-pool classifier head.  Shares the
TP-sharded attention/MLP blocks of the model zoo; the encoder runs on the
same gfx950 kernels as BERT.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..mesh import DeviceMesh
from ..parallel.layers import tag_seed
from .bert import BertAttention
from .gpt import LayerNorm, MLP


@dataclass
class ViTConfig:
    image_size: int = 224
    patch_size: int = 16
    num_channels: int = 3
    hidden_size: int = 768
    num_layers: int = 12
    num_heads: int = 12
    num_classes: int = 1000
    ffn_mult: int = 4
    layernorm_eps: float = 1e-6

    @property
    def seq_len(self) -> int:
        return (self.image_size // self.patch_size) ** 2


class ViTBlock(nn.Module):
    """Pre-LN encoder block (ViT convention)."""

    def __init__(self, cfg, mesh, axis, dtype, device, layer_idx,
                 init_seed):
        super().__init__()
        self.ln1 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                             device)
        self.attn = BertAttention(cfg, mesh, axis, dtype, device,
                                  layer_idx, init_seed)
        self.ln2 = LayerNorm(cfg.hidden_size, cfg.layernorm_eps, dtype,
                             device)
        self.mlp = MLP(cfg, mesh, axis, dtype, device, layer_idx,
                       init_seed)

    def forward(self, x):
        x = x + self.attn(self.ln1(x))
        x = x + self.mlp(self.ln2(x))
        return x


class ViTModel(nn.Module):

    def __init__(self, cfg: ViTConfig, mesh: Optional[DeviceMesh] = None,
                 axis: int = 1, dtype=torch.float32, device=None,
                 init_seed: int = 0):
        super().__init__()
        self.cfg = cfg
        P, H = cfg.patch_size, cfg.hidden_size

        def init_p(shape, tag, std=0.02):
            gen_dev = device if (device is not None and
                                 torch.device(device).type == "cuda") \
                else "cpu"
            g = torch.Generator(device=gen_dev)
            g.manual_seed(tag_seed(init_seed, tag))
            w = torch.empty(shape, dtype=torch.float32, device=gen_dev)
            w.normal_(0, std, generator=g)
            return nn.Parameter(w.to(dtype=dtype, device=device))

        # patch embedding as a linear over flattened patches (equivalent
        # to the stride-P conv, but lands on hipBLASLt as one GEMM)
        self.patch_proj = init_p((cfg.num_channels * P * P, H),
                                 "vit.patch")
        self.patch_bias = nn.Parameter(torch.zeros(H, dtype=dtype,
                                                   device=device))
        self.pos_emb = init_p((cfg.seq_len, H), "vit.pos")
        self.blocks = nn.ModuleList([
            ViTBlock(cfg, mesh, axis, dtype, device, i, init_seed)
            for i in range(cfg.num_layers)
        ])
        self.ln_f = LayerNorm(H, cfg.layernorm_eps, dtype, device)
        self.head_w = init_p((H, cfg.num_classes), "vit.head")
        self.head_b = nn.Parameter(torch.zeros(cfg.num_classes,
                                               dtype=dtype, device=device))

    def _patchify(self, images: torch.Tensor) -> torch.Tensor:
        """[B, C, H, W] -> [B, S, C*P*P] flattened patches."""
        B, C, Hh, W = images.shape
        P = self.cfg.patch_size
        x = images.reshape(B, C, Hh // P, P, W // P, P)
        return x.permute(0, 2, 4, 1, 3, 5).reshape(B, -1, C * P * P)

    def forward(self, images: torch.Tensor) -> torch.Tensor:
        """images [B, C, H, W] -> logits [B, num_classes]."""
        x = self._patchify(images) @ self.patch_proj + self.patch_bias
        x = x + self.pos_emb
        for blk in self.blocks:
            x = blk(x)
        x = self.ln_f(x).mean(dim=1)
        return x @ self.head_w + self.head_b

    def loss(self, images: torch.Tensor,
             labels: torch.Tensor) -> torch.Tensor:
        return torch.nn.functional.cross_entropy(
            self.forward(images).float(), labels)
