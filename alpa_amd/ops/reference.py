"""Pure-PyTorch reference implementations of every hot op.

These serve two purposes:
1. CPU execution path (GPU-free unit tests, gloo multi-process tests).
2. Numerics oracle for the HIP kernels (tests compare the gfx950 kernel
   against these in fp32; see tests/test_kernels_gpu.py).

They are *not* used on a GPU box — `alpa_amd.ops` raises if the HIP extension
is missing there (anti-silent-fallback), unless explicitly overridden.
"""
from __future__ import annotations

import math
from typing import Tuple

import torch
import torch.nn.functional as F



def _up(t: torch.Tensor) -> torch.Tensor:
    """Upcast half-precision to fp32 for math; leave fp32/fp64 alone."""
    if t.dtype in (torch.bfloat16, torch.float16):
        return t.float()
    return t

def layer_norm_fwd(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
                   eps: float) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (y, mean, rstd). mean/rstd are fp32 per-row stats."""
    xf = _up(x)
    mean = xf.mean(dim=-1)
    var = xf.var(dim=-1, unbiased=False)
    rstd = torch.rsqrt(var + eps)
    y = (xf - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
    y = y * _up(weight) + _up(bias)
    return y.to(x.dtype), mean, rstd


def layer_norm_bwd(dy: torch.Tensor, x: torch.Tensor, weight: torch.Tensor,
                   mean: torch.Tensor, rstd: torch.Tensor
                   ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (dx, dweight, dbias)."""
    xf = _up(x)
    dyf = _up(dy)
    xhat = (xf - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
    wdy = dyf * _up(weight)
    H = x.shape[-1]
    c1 = wdy.mean(dim=-1, keepdim=True)
    c2 = (wdy * xhat).mean(dim=-1, keepdim=True)
    dx = (wdy - c1 - xhat * c2) * rstd.unsqueeze(-1)
    dims = tuple(range(x.dim() - 1))
    dweight = (dyf * xhat).sum(dim=dims)
    dbias = dyf.sum(dim=dims)
    return dx.to(x.dtype), dweight, dbias


def rms_norm_fwd(x: torch.Tensor, weight: torch.Tensor,
                 eps: float) -> Tuple[torch.Tensor, torch.Tensor]:
    xf = _up(x)
    rstd = torch.rsqrt(xf.pow(2).mean(dim=-1) + eps)
    y = xf * rstd.unsqueeze(-1) * _up(weight)
    return y.to(x.dtype), rstd


def rms_norm_bwd(dy: torch.Tensor, x: torch.Tensor, weight: torch.Tensor,
                 rstd: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    xf = _up(x)
    dyf = _up(dy)
    xhat = xf * rstd.unsqueeze(-1)
    wdy = dyf * _up(weight)
    H = x.shape[-1]
    c = (wdy * xhat).mean(dim=-1, keepdim=True)
    dx = (wdy - xhat * c) * rstd.unsqueeze(-1)
    dims = tuple(range(x.dim() - 1))
    dweight = (dyf * xhat).sum(dim=dims)
    return dx.to(x.dtype), dweight


def gelu(x: torch.Tensor) -> torch.Tensor:
    # tanh approximation — matches the HIP kernel exactly
    return 0.5 * x * (1.0 + torch.tanh(0.7978845608028654 *
                                       (x + 0.044715 * x * x * x)))


def gelu_grad(x: torch.Tensor) -> torch.Tensor:
    k = 0.7978845608028654
    x3 = x * x * x
    t = torch.tanh(k * (x + 0.044715 * x3))
    dt = (1.0 - t * t) * k * (1.0 + 3 * 0.044715 * x * x)
    return 0.5 * (1.0 + t) + 0.5 * x * dt


def bias_gelu_fwd(x: torch.Tensor, bias: torch.Tensor) -> torch.Tensor:
    return gelu((_up(x) + _up(bias))).to(x.dtype)


def bias_gelu_bwd(dy: torch.Tensor, x: torch.Tensor, bias: torch.Tensor
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    z = _up(x) + _up(bias)
    dx = _up(dy) * gelu_grad(z)
    dims = tuple(range(x.dim() - 1))
    dbias = dx.sum(dim=dims)
    return dx.to(x.dtype), dbias


def _alibi_bias(alibi, Hh, S, Skv, device):
    """[1, Hh, S, Skv] ALiBi bias: slope_h * (kv_pos - q_pos); q global
    position is offset by Skv - S (cached decode)."""
    qpos = torch.arange(S, device=device, dtype=torch.float32) + (Skv - S)
    kpos = torch.arange(Skv, device=device, dtype=torch.float32)
    rel = kpos.view(1, 1, 1, Skv) - qpos.view(1, 1, S, 1)
    return alibi.float().view(1, Hh, 1, 1) * rel


def attention_fwd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  causal: bool = True, softmax_scale: float | None = None,
                  alibi: torch.Tensor | None = None,
                  kv_lens: torch.Tensor | None = None
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    """q,k,v: [B, Hh, S, D]. Returns (o, lse[B,Hh,S]) in fp32 math.
    alibi: optional per-head slopes [Hh] (BLOOM bias).
    kv_lens: optional int [B] per-batch real KV lengths (varlen decode:
    positions >= kv_lens[b] are masked; alibi q position offsets per
    batch)."""
    B, Hh, S, D = q.shape
    scale = softmax_scale if softmax_scale is not None else 1.0 / math.sqrt(D)
    s = torch.matmul(_up(q), _up(k).transpose(-1, -2)) * scale
    Skv = k.shape[2]
    if alibi is not None:
        if kv_lens is not None:
            kpos = torch.arange(Skv, device=q.device, dtype=torch.float32)
            qpos = (kv_lens.to(q.device).float().view(B, 1, 1, 1) - S +
                    torch.arange(S, device=q.device,
                                 dtype=torch.float32).view(1, 1, S, 1))
            s = s + alibi.float().view(1, Hh, 1, 1) * (
                kpos.view(1, 1, 1, Skv) - qpos)
        else:
            s = s + _alibi_bias(alibi, Hh, S, Skv, q.device)
    if kv_lens is not None:
        mask = torch.arange(Skv, device=q.device).view(1, 1, 1, Skv) >=             kv_lens.to(q.device).view(B, 1, 1, 1)
        s = s.masked_fill(mask, float("-inf"))
    if causal:
        mask = torch.triu(torch.ones(S, k.shape[2], dtype=torch.bool,
                                     device=q.device), diagonal=1)
        s = s.masked_fill(mask, float("-inf"))
    m = s.max(dim=-1, keepdim=True).values
    p = torch.exp(s - m)
    l = p.sum(dim=-1, keepdim=True)
    o = torch.matmul(p / l, _up(v))
    lse = (m + torch.log(l)).squeeze(-1)
    return o.to(q.dtype), lse


def attention_bwd(do: torch.Tensor, q: torch.Tensor, k: torch.Tensor,
                  v: torch.Tensor, o: torch.Tensor, lse: torch.Tensor,
                  causal: bool = True, softmax_scale: float | None = None,
                  alibi: torch.Tensor | None = None
                  ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    B, Hh, S, D = q.shape
    scale = softmax_scale if softmax_scale is not None else 1.0 / math.sqrt(D)
    qf, kf, vf, dof = _up(q), _up(k), _up(v), _up(do)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if alibi is not None:
        s = s + _alibi_bias(alibi, Hh, S, k.shape[2], q.device)
    if causal:
        mask = torch.triu(torch.ones(S, k.shape[2], dtype=torch.bool,
                                     device=q.device), diagonal=1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.exp(s - lse.unsqueeze(-1))
    dv = torch.matmul(p.transpose(-1, -2), dof)
    dp = torch.matmul(dof, vf.transpose(-1, -2))
    delta = (dof * _up(o)).sum(dim=-1, keepdim=True)
    ds = p * (dp - delta) * scale
    dq = torch.matmul(ds, kf)
    dk = torch.matmul(ds.transpose(-1, -2), qf)
    return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype)


def softmax_cross_entropy_fwd(logits: torch.Tensor, targets: torch.Tensor
                              ) -> Tuple[torch.Tensor, torch.Tensor]:
    """logits [N, V], targets [N] int64. Returns (loss[N], lse[N])."""
    lf = _up(logits)
    lse = torch.logsumexp(lf, dim=-1)
    loss = lse - lf.gather(-1, targets.unsqueeze(-1)).squeeze(-1)
    return loss, lse


def softmax_cross_entropy_bwd(dloss: torch.Tensor, logits: torch.Tensor,
                              targets: torch.Tensor, lse: torch.Tensor
                              ) -> torch.Tensor:
    p = torch.exp(_up(logits) - lse.unsqueeze(-1))
    p.scatter_add_(-1, targets.unsqueeze(-1),
                   -torch.ones_like(targets, dtype=p.dtype).unsqueeze(-1))
    return (p * _up(dloss).unsqueeze(-1)).to(logits.dtype)


def adamw_step(params, grads, exp_avgs, exp_avg_sqs, step: int, lr: float,
               beta1: float, beta2: float, eps: float, weight_decay: float,
               grad_scale: float = 1.0):
    """In-place multi-tensor AdamW (fp32 master math on fp32 state)."""
    bc1 = 1.0 - beta1 ** step
    bc2 = 1.0 - beta2 ** step
    for p, g, m, v in zip(params, grads, exp_avgs, exp_avg_sqs):
        gf = _up(g) * grad_scale
        pf = _up(p)
        pf.mul_(1.0 - lr * weight_decay)
        m.mul_(beta1).add_(gf, alpha=1.0 - beta1)
        v.mul_(beta2).addcmul_(gf, gf, value=1.0 - beta2)
        denom = (v / bc2).sqrt_().add_(eps)
        pf.addcdiv_(m / bc1, denom, value=-lr)
        p.copy_(pf.to(p.dtype))
