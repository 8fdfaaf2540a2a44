"""fp8 (e4m3) forward GEMMs — EXPERIMENTAL, off by default.

MI355X's matrix cores run fp8 at 2x the bf16 rate (≈5 PF dense); the
GEMM-bound ~55% of the training step is the target.  This module
implements the standard "fp8 forward, bf16 backward" recipe: the forward
GEMM quantizes X and W to float8_e4m3fn with per-tensor amax scaling and
runs `torch._scaled_mm` (hipBLASLt fp8 under ROCm — probed working on
this stack, ~3.6% rel err); the backward keeps exact bf16 GEMMs, so
gradients match the bf16 path up to the forward quantization error.

NOT used by the flagship benchmark: BASELINE comparisons require bf16
compute.  Enable per-layer via global_config.fp8_gemm
(`ALPA_AMD_FP8=1`) for experiments; a delayed-scaling recipe and fp8
backward are round-2 work.
"""
from __future__ import annotations

from typing import Optional

import torch

E4M3_MAX = 448.0

# epoch counter: optimizers bump it after each parameter update so the
# per-module quantized-weight caches invalidate (the fused AdamW kernel
# writes through raw pointers and does not touch torch's version
# counters)
_EPOCH = 0


def bump_epoch() -> None:
    global _EPOCH
    _EPOCH += 1


def quantize_weight_cached(module, weight: torch.Tensor):
    """(wq, ws) for `module.weight`, re-quantized once per optimizer
    epoch instead of per forward call — the weight-side half of the
    naive recipe's per-GEMM amax+cast overhead disappears."""
    cached = getattr(module, "_fp8_cache", None)
    if cached is not None and cached[0] == _EPOCH:
        return cached[1], cached[2]
    wq, ws = _quantize(weight)
    module._fp8_cache = (_EPOCH, wq, ws)
    return wq, ws


def _quantize(t: torch.Tensor):
    """Per-tensor symmetric scaling into float8_e4m3fn; returns
    (fp8 tensor, fp32 scale such that t ≈ t_fp8 * scale)."""
    amax = t.abs().amax().clamp_min(1e-12).float()
    scale = amax / E4M3_MAX
    q = (t.float() / scale).clamp(-E4M3_MAX, E4M3_MAX).to(
        torch.float8_e4m3fn)
    return q, scale


def fp8_available(x: torch.Tensor) -> bool:
    return (x.is_cuda and hasattr(torch, "float8_e4m3fn")
            and hasattr(torch, "_scaled_mm"))


class _Fp8Linear(torch.autograd.Function):
    """y = x @ w.T (+ bias): fp8 forward GEMM, exact bf16 backward."""

    @staticmethod
    def forward(ctx, x, w, bias, wq=None, ws=None):
        shape = x.shape
        x2 = x.reshape(-1, shape[-1])
        xq, xs = _quantize(x2)
        if wq is None:
            wq, ws = _quantize(w)
        # _scaled_mm wants B column-major: w.T with w row-major qualifies
        y = torch._scaled_mm(xq, wq.t(), scale_a=xs, scale_b=ws,
                             bias=bias, out_dtype=x.dtype)
        ctx.save_for_backward(x2, w)
        ctx.has_bias = bias is not None
        return y.reshape(*shape[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2, w = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        dx = (dy2 @ w).reshape(*dy.shape[:-1], w.shape[1])
        dw = dy2.t() @ x2
        db = dy2.sum(0) if ctx.has_bias else None
        return dx, dw, db, None, None


def fp8_linear(x: torch.Tensor, weight: torch.Tensor,
               bias: Optional[torch.Tensor] = None,
               module=None) -> torch.Tensor:
    """Drop-in for F.linear with an fp8 forward GEMM.  Requires CUDA and
    dims divisible by 16 (hipBLASLt fp8 tile constraint); callers gate on
    `fp8_available` and fall back to bf16 matmul otherwise.  Passing the
    owning `module` enables the per-epoch quantized-weight cache."""
    if module is not None:
        wq, ws = quantize_weight_cached(module, weight)
        return _Fp8Linear.apply(x, weight, bias, wq, ws)
    return _Fp8Linear.apply(x, weight, bias)
