"""Is the fp8 _scaled_mm competitive at decode shapes (small M)?
Times bf16 matmul vs fp8 scaled_mm for M in {16..256} at OPT-13B layer
shapes.  Decode is weight-bandwidth-bound: fp8 halves the bytes, but
only if hipBLASLt has a skinny-M fp8 path."""
import sys
import time

import torch

torch.manual_seed(0)
dev = "cuda"
K = N = 5120
for M in (16, 32, 64, 128, 256):
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    xq = x.to(torch.float8_e4m3fn)
    wq = w.to(torch.float8_e4m3fn)
    one = torch.ones(1, device=dev)

    def bench(fn, iters=200):
        for _ in range(20):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1e6

    t_bf = bench(lambda: x @ w.t())
    t_f8 = bench(lambda: torch._scaled_mm(xq, wq.t(), scale_a=one,
                                          scale_b=one,
                                          out_dtype=torch.bfloat16))
    print(f"M={M:4d}: bf16 {t_bf:7.1f} us   fp8 {t_f8:7.1f} us   "
          f"ratio {t_bf / t_f8:.2f}x")
