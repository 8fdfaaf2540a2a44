"""2-D UNet (diffusion-style; reference ``alpa/model/unet_2d.py``,
benchmarked in suite_unet.py).  Compact encoder/decoder with timestep
embedding and mid-block self-attention through the flash kernel.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops


def timestep_embedding(t: torch.Tensor, dim: int) -> torch.Tensor:
    half = dim // 2
    freqs = torch.exp(-math.log(10000) *
                      torch.arange(half, dtype=torch.float32,
                                   device=t.device) / half)
    args = t.float().unsqueeze(-1) * freqs
    return torch.cat([torch.cos(args), torch.sin(args)], dim=-1)


class ResBlock(nn.Module):

    def __init__(self, in_ch, out_ch, temb_ch):
        super().__init__()
        self.norm1 = nn.GroupNorm(8, in_ch)
        self.conv1 = nn.Conv2d(in_ch, out_ch, 3, padding=1)
        self.temb = nn.Linear(temb_ch, out_ch)
        self.norm2 = nn.GroupNorm(8, out_ch)
        self.conv2 = nn.Conv2d(out_ch, out_ch, 3, padding=1)
        self.skip = nn.Conv2d(in_ch, out_ch, 1) if in_ch != out_ch \
            else nn.Identity()

    def forward(self, x, temb):
        h = self.conv1(F.silu(self.norm1(x)))
        h = h + self.temb(F.silu(temb))[:, :, None, None]
        h = self.conv2(F.silu(self.norm2(h)))
        return h + self.skip(x)


class SelfAttention2d(nn.Module):
    """Spatial self-attention via the flash kernel (HW x HW)."""

    def __init__(self, ch, heads=None):
        super().__init__()
        self.norm = nn.GroupNorm(8, ch)
        self.qkv = nn.Linear(ch, 3 * ch)
        self.proj = nn.Linear(ch, ch)
        # flash kernel wants head_dim multiple of 16
        self.heads = heads if heads is not None else max(1, ch // 32)
        assert (ch // self.heads) % 16 == 0, (ch, self.heads)

    def forward(self, x):
        B, C, H, W = x.shape
        h = self.norm(x).flatten(2).transpose(1, 2)  # [B, HW, C]
        qkv = self.qkv(h).view(B, H * W, self.heads, 3, C // self.heads)
        qkv = qkv.view(B, H * W, -1)
        o = ops.flash_attention_qkv(qkv, self.heads, causal=False)
        o = self.proj(o)
        return x + o.transpose(1, 2).view(B, C, H, W)


class UNet2D(nn.Module):

    def __init__(self, in_ch: int = 3, base: int = 32,
                 ch_mults=(1, 2, 4), dtype=torch.float32, device=None):
        super().__init__()
        temb_ch = base * 4
        self.temb = nn.Sequential(nn.Linear(base, temb_ch), nn.SiLU(),
                                  nn.Linear(temb_ch, temb_ch))
        self.base = base
        self.conv_in = nn.Conv2d(in_ch, base, 3, padding=1)
        chs = [base * m for m in ch_mults]
        self.down = nn.ModuleList()
        cur = base
        for ch in chs:
            self.down.append(ResBlock(cur, ch, temb_ch))
            cur = ch
        self.downsample = nn.ModuleList(
            [nn.Conv2d(ch, ch, 3, stride=2, padding=1) for ch in chs])
        self.mid1 = ResBlock(cur, cur, temb_ch)
        self.mid_attn = SelfAttention2d(cur)
        self.mid2 = ResBlock(cur, cur, temb_ch)
        self.up = nn.ModuleList()
        self.upsample = nn.ModuleList()
        for ch in reversed(chs):
            self.upsample.append(nn.ConvTranspose2d(cur, ch, 4, stride=2,
                                                    padding=1))
            self.up.append(ResBlock(ch * 2, ch, temb_ch))
            cur = ch
        self.norm_out = nn.GroupNorm(8, cur)
        self.conv_out = nn.Conv2d(cur, in_ch, 3, padding=1)
        self.to(dtype=dtype)
        if device is not None:
            self.to(device)

    def forward(self, x, t):
        temb = self.temb(timestep_embedding(t, self.base).to(x.dtype))
        h = self.conv_in(x)
        skips = []
        for blk, ds in zip(self.down, self.downsample):
            h = blk(h, temb)
            skips.append(h)
            h = ds(h)
        h = self.mid2(self.mid_attn(self.mid1(h, temb)), temb)
        for blk, us in zip(self.up, self.upsample):
            h = us(h)
            h = blk(torch.cat([h, skips.pop()], dim=1), temb)
        return self.conv_out(F.silu(self.norm_out(h)))

    def loss(self, x, t, noise):
        """Denoising MSE (diffusion training objective)."""
        return F.mse_loss(self.forward(x, t), noise)
