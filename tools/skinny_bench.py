"""Numerics + bandwidth of the skinny decode GEMV (csrc/skinny_gemm.hip)
vs torch.matmul at OPT-13B decode shapes."""
import sys
import time

import torch

sys.path.insert(0, ".")
from alpa_amd.ops import _skinny_splits
from alpa_amd.ops._backend import hip_ops


def run(M, N, K, fp8):
    x = (torch.randn(M, K, device="cuda") * 0.5).to(torch.bfloat16)
    w = (torch.randn(N, K, device="cuda") * 0.05).to(torch.bfloat16)
    if fp8:
        amax = w.abs().amax().float().clamp_min(1e-12)
        scale = (amax / 448.0).reshape(1)
        q = (w.float() / scale).clamp(-448, 448).to(torch.float8_e4m3fn)
        wp = q.view(torch.uint8).view(N, K // 8, 8).permute(1, 0, 2) \
            .contiguous()
        wref = (q.float() * scale).to(torch.bfloat16)
    else:
        scale = None
        wp = w.view(N, K // 8, 8).permute(1, 0, 2).contiguous()
        wref = w
    import os
    s = _skinny_splits(N, K, M)
    sweep = os.environ.get("SKINNY_SWEEP") == "1"
    y = hip_ops().skinny_gemm(wp, x, scale, None, N, K, s)
    ref = x.float() @ wref.float().t()
    rel = (y - ref).abs().mean() / ref.abs().mean().clamp_min(1e-9)

    def bench(fn, iters=300):
        for _ in range(30):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1e6

    t_k = bench(lambda: hip_ops().skinny_gemm(wp, x, scale, None, N, K, s))
    if sweep:
        q = K // 64
        MT = 4
        while MT < M:
            MT *= 2
        for s2 in [d for d in range(1, q + 1) if q % d == 0
                   and (K // 8 // d) * 16 * MT <= 65536]:
            t2 = bench(lambda: hip_ops().skinny_gemm(wp, x, scale, None, N, K,
                                                     s2), 100)
            if t2 < t_k:
                t_k, s = t2, s2
    t_t = bench(lambda: x @ w.t())
    bytes_w = N * K * (1 if fp8 else 2)
    print(f"M={M:3d} N={N:6d} K={K:6d} {'fp8' if fp8 else 'bf16'} "
          f"splits={s:2d}: rel={rel.item():.4f}  kernel {t_k:7.1f} us "
          f"({bytes_w / t_k / 1e6:5.2f} TB/s)  torch {t_t:7.1f} us  "
          f"speedup {t_t / t_k:4.2f}x")


import os as _os

_shapes = _os.environ.get("SKINNY_SHAPES")
SHAPES = ([tuple(map(int, t.split("x"))) for t in _shapes.split(",")]
          if _shapes else [(5120, 5120), (15360, 5120), (20480, 5120),
                           (5120, 20480)])
MS = tuple(int(m) for m in
           _os.environ.get("SKINNY_MS", "4,16,64").split(","))
for fp8 in (False, True):
    for (N, K) in SHAPES:
        for M in MS:
            run(M, N, K, fp8)
