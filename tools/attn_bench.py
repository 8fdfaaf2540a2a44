"""Isolated attention kernel microbenchmark (bench shapes).

Usage: python tools/attn_bench.py [--iters 20]
Prints per-kernel time + effective TFLOPS for fwd / bwd on the GPT-2.6B
bench shape (B=8, h=32, S=1024, D=80).
"""
import argparse
import math
import sys
import time

import torch

import os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
from alpa_amd.ops._backend import hip_ops


def timeit(fn, iters):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--ablate", action="store_true")
    p.add_argument("--B", type=int, default=8)
    p.add_argument("--H", type=int, default=32)
    p.add_argument("--S", type=int, default=1024)
    p.add_argument("--D", type=int, default=80)
    args = p.parse_args()
    ext = hip_ops()
    B, H, S, D = args.B, args.H, args.S, args.D
    torch.manual_seed(0)
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    scale = 1.0 / math.sqrt(D)

    fwd_flops = 4 * B * H * S * S * D * 0.5  # causal
    t = timeit(lambda: ext.attn_fwd(q, k, v, True, scale), args.iters)
    print(f"fwd:  {t*1e6:8.1f} us  {fwd_flops/t/1e12:7.1f} TF")

    if hasattr(ext, "attn_fwd_sbuf"):
        o1, l1 = ext.attn_fwd(q, k, v, True, scale)
        o3, l3 = ext.attn_fwd_sbuf(q, k, v, True, scale)
        err = (o1.float() - o3.float()).abs().max().item()
        t = timeit(lambda: ext.attn_fwd_sbuf(q, k, v, True, scale),
                   args.iters)
        print(f"sbuf: {t*1e6:8.1f} us  {fwd_flops/t/1e12:7.1f} TF  "
              f"(maxdiff vs db: {err:.4f})")

    if hasattr(ext, "attn_fwd_v2"):
        o1, l1 = ext.attn_fwd(q, k, v, True, scale)
        o2, l2 = ext.attn_fwd_v2(q, k, v, True, scale)
        err = (o1.float() - o2.float()).abs().max().item()
        t = timeit(lambda: ext.attn_fwd_v2(q, k, v, True, scale),
                   args.iters)
        print(f"fwd32:{t*1e6:8.1f} us  {fwd_flops/t/1e12:7.1f} TF  "
              f"(maxdiff vs v1: {err:.4f})")

    if args.ablate:
        names = {0: "full", 1: "noPV", 2: "noSM(QK+pwrite)",
                 3: "QKonly", 4: "staging"}
        for abl in (0, 1, 2, 3, 4):
            t = timeit(lambda: ext.attn_fwd_ablate(q, k, v, True, scale,
                                                   abl), args.iters)
            print(f"  abl{abl} {names[abl]:16s} {t*1e6:8.1f} us")

    o, lse = ext.attn_fwd(q, k, v, True, scale)
    do = torch.randn_like(o)
    lse3 = lse.view(B, H, S)
    bwd_flops = 10 * B * H * S * S * D * 0.5
    t = timeit(lambda: ext.attn_bwd(do, q, k, v, o, lse3, True, scale),
               args.iters)
    print(f"bwd:  {t*1e6:8.1f} us  {bwd_flops/t/1e12:7.1f} TF")

    if args.ablate:
        delta = (do.float() * o.float()).sum(-1)
        dk = torch.empty_like(k)
        dv = torch.empty_like(v)
        names = {0: "dkv full", 1: "no dK", 2: "S/dP+math", 3: "staging"}
        for abl in (0, 1, 2, 3):
            t = timeit(lambda: ext.attn_bwd_dkv_ablate(
                do, q, k, v, lse3, delta, dk, dv, True, scale, abl),
                args.iters)
            print(f"  dkv abl{abl} {names[abl]:12s} {t*1e6:8.1f} us")


if __name__ == "__main__":
    main()
