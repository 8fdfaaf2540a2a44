"""fp8 quantize kernel microbenchmark: GB/s vs the HBM roofline."""
import sys, time
sys.path.insert(0, ".")
import torch
from alpa_amd.ops._backend import hip_ops

def t(fn, it=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / it

ext = hip_ops()
for (M, N) in [(32768, 2560), (32768, 7680), (32768, 10240)]:
    x = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    sc = torch.tensor([0.01], device="cuda")
    dt_s = t(lambda: ext.fp8_quantize(x, sc, False, False))
    dt_d = t(lambda: ext.fp8_quantize(x, sc, True, False))
    gb_s = (M*N*3) / dt_s / 1e9   # 2B read + 1B write
    gb_d = (M*N*4) / dt_d / 1e9   # 2B read + 2x1B write
    print(f"[{M}x{N}] single {dt_s*1e6:7.1f}us {gb_s:6.0f} GB/s   "
          f"dual {dt_d*1e6:7.1f}us {gb_d:6.0f} GB/s")
