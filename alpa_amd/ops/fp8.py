"""fp8 (OCP e4m3) GEMM recipe for gfx950 — forward AND backward.

MI355X matrix cores run fp8 at ~2x the bf16 rate; measured at the
GPT-2.6B step shapes (profiles/fp8_bench): 2.6-3.1 PF vs 1.35-1.55 PF
bf16.  Round 1 lost the win to cast overhead (torch-op amax+cast up to
7.5 ms/tensor); round 2 fixes that with one-pass HIP quantize kernels
(ops/csrc/fp8_quant.hip) and quantizes all three GEMMs:

  fwd : y  = Xq  @ Wq^T         (X e4m3 via fused one-pass quantize)
  bwd : dX = dYq @ Wq           (dY e4m3, W from the per-step cache)
        dW = dYq^T @ Xq         (transposed twins from DUAL quantize —
                                 no extra HBM pass for either operand)

Scaling is DELAYED (Transformer-Engine style): each tensor role keeps a
running amax on device (`r = max(amax_now, 0.999 r)`) and the NEXT
quantize uses scale = r / 448.  Everything is device-side in-place, so
the whole recipe is hipGraph-capturable: on replay the captured
quantize kernels re-read the updated weights/running-amax in place.

Numerics: e4m3 everywhere with per-tensor delayed scaling (the
DeepSeek-V3 style choice); the loss-curve check against bf16 lives in
tools/fp8_losscheck.py with results in docs/BENCHMARK.md.

Enable via global_config.fp8_gemm (ALPA_AMD_FP8=1).  The flagship
BASELINE bench stays bf16 by default; `bench.py --fp8` reports the fp8
number separately.
"""
from __future__ import annotations

from typing import Optional

import torch

from ._backend import hip_ops

E4M3_MAX = 448.0
AMAX_DECAY = 0.999

# epoch counter: optimizers bump it after each parameter update so
# EAGER-mode per-module quantized-weight caches invalidate.  Under
# hipGraph capture the quantize kernels are captured in-step (the cache
# misses during capture) and re-run on every replay, reading the updated
# weights in place — no epoch check executes at replay time.
_EPOCH = 0


def bump_epoch() -> None:
    global _EPOCH
    _EPOCH += 1


def fp8_available(x: torch.Tensor) -> bool:
    return (x.is_cuda and hasattr(torch, "float8_e4m3fn")
            and hasattr(torch, "_scaled_mm"))


class _RoleState:
    """Delayed-scaling state for one tensor role (module activation /
    grad): persistent device tensors, updated in place (graph-safe)."""

    __slots__ = ("running", "scale")

    def __init__(self):
        self.running = None
        self.scale = None

    def ensure(self, ref: torch.Tensor):
        if self.scale is None:
            with torch.no_grad():
                amax = ref.abs().amax().float().clamp_min(1e-12).reshape(1)
                self.running = amax.clone()
                self.scale = (amax / E4M3_MAX).contiguous()

    @torch.no_grad()
    def update(self, amax: torch.Tensor):
        # r <- max(amax, decay * r); scale <- r / 448, all in place
        torch.maximum(amax, self.running * AMAX_DECAY, out=self.running)
        self.scale.copy_(self.running.clamp_min(1e-12) / E4M3_MAX)


def _role(module, name: str) -> _RoleState:
    st = getattr(module, f"_fp8_{name}", None)
    if st is None:
        st = _RoleState()
        setattr(module, f"_fp8_{name}", st)
    return st


def quantize(x2: torch.Tensor, state: _RoleState, dual: bool):
    """One-pass delayed-scale quantize (+ transposed twin when dual).
    Returns (q, qt|None, used_scale); updates the role's running amax.
    `used_scale` is a snapshot — state.scale mutates in place right
    after, and the dequant scale must match the cast actually done."""
    state.ensure(x2)
    used = state.scale.clone()
    q, qt, amax = hip_ops().fp8_quantize(x2, used, dual, False)
    state.update(amax)
    return q, qt, used


def quantize_weight_cached(module, weight: torch.Tensor,
                           dual: bool = True):
    """(wq [n,k], wqt [k,n], ws) re-quantized once per optimizer epoch in
    eager mode; captured in-step under hipGraphs (see module doc).
    Weight scaling is delayed like activations (running amax from the
    kernel's own one-pass reduction — weights drift slowly per step, and
    the separate torch abs+amax passes cost ~2% of the step in the
    r2 profile)."""
    cached = getattr(module, "_fp8_cache", None)
    graphing = torch.cuda.is_current_stream_capturing() \
        if weight.is_cuda else False
    if cached is not None and cached[0] == _EPOCH and not graphing \
            and (not dual or cached[2] is not None):
        return cached[1], cached[2], cached[3]
    st = _role(module, "w")
    with torch.no_grad():
        wq, wqt, ws = quantize(weight.contiguous(), st, dual=dual)
    module._fp8_cache = (_EPOCH, wq, wqt, ws)
    return wq, wqt, ws


class _Fp8Linear(torch.autograd.Function):
    """y = x @ w.T (+ bias) with fp8 fwd AND bwd GEMMs (see module doc).
    Saves the fp8 transposed activation instead of the bf16 input —
    halves the saved-activation bytes as a side effect."""

    @staticmethod
    def forward(ctx, x, w, bias, module, infer):
        # `infer` is computed by the fp8_linear wrapper: grad mode is
        # ALWAYS disabled inside Function.forward, so is_grad_enabled()
        # here would misread training as inference.
        from ..global_env import global_config
        shape = x.shape
        x2 = x.reshape(-1, shape[-1]).contiguous()
        sx = _role(module, "x")
        wgrad_fp8 = global_config.fp8_wgrad
        dx_bf16 = global_config.fp8_dx_bf16
        xq, xqt, x_scale = quantize(x2, sx, dual=wgrad_fp8 and not infer)
        wq, wqt, ws = quantize_weight_cached(
            module, w, dual=(not dx_bf16) and not infer)
        y = torch._scaled_mm(xq, wq.t(), scale_a=x_scale, scale_b=ws,
                             bias=bias, out_dtype=x.dtype)
        if not infer:
            ctx.module = module
            ctx.wgrad_fp8 = wgrad_fp8
            ctx.dx_bf16 = dx_bf16
            wb = w if dx_bf16 else wqt
            if wgrad_fp8:
                ctx.save_for_backward(xqt, wb, x_scale, ws)
            else:
                ctx.save_for_backward(x2, wb, x_scale, ws)
            ctx.has_bias = bias is not None
        return y.reshape(*shape[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        xsaved, wb, sx, ws = ctx.saved_tensors
        module = ctx.module
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        if ctx.wgrad_fp8 or not ctx.dx_bf16:
            sg = _role(module, "g")
            dyq, dyqt, g_scale = quantize(dy2, sg, dual=ctx.wgrad_fp8)
        if ctx.dx_bf16:
            # huge-model mode: dX on the bf16 master weight (no wqt
            # cache — halves the fp8 weight footprint)
            dx = dy2 @ wb
        else:
            # dX [M,k] = dY [M,n] @ W [n,k]; B col-major = wqt.t()
            dx = torch._scaled_mm(dyq, wb.t(), scale_a=g_scale,
                                  scale_b=ws, out_dtype=dy.dtype)
        if ctx.wgrad_fp8:
            # dW [n,k] = dY^T [n,M] @ X [M,k]; A row-major = dyqt,
            # B col-major = xqt.t()
            dw = torch._scaled_mm(dyqt, xsaved.t(), scale_a=g_scale,
                                  scale_b=sx, out_dtype=dy.dtype)
        else:
            dw = dy2.t() @ xsaved     # exact bf16 weight grad
        db = None
        if ctx.has_bias:
            db = hip_ops().colsum_bf16(dy2).to(dy.dtype)
        k = wb.shape[1] if ctx.dx_bf16 else wb.shape[0]
        return (dx.reshape(*dy.shape[:-1], k), dw, db, None, None)


def fp8_linear(x: torch.Tensor, weight: torch.Tensor,
               bias: Optional[torch.Tensor] = None,
               module=None) -> torch.Tensor:
    """Drop-in for F.linear with fp8 fwd+bwd GEMMs.  Requires CUDA and
    dims divisible by 16 (hipBLASLt fp8 tiles); callers gate on
    `fp8_available`.  `module` anchors the per-role delayed-scaling
    state and the per-epoch weight cache."""
    assert module is not None, "fp8_linear needs the owning module"
    # inference (serving decode/prefill under no_grad): no backward
    # operands — single-layout quantize for both x and w, and the
    # per-module weight cache holds ONE fp8 copy read at half the
    # bf16 bytes/step (decode is weight-bandwidth-bound)
    infer = not torch.is_grad_enabled()
    return _Fp8Linear.apply(x, weight, bias, module, infer)
