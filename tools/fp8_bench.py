"""fp8 vs bf16 GEMM microbenchmark at the GPT-2.6B step shapes.

Decides the round-2 fp8 recipe with data: measures
torch._scaled_mm (hipBLASLt fp8) against bf16 torch.matmul for the
forward and backward GEMM shapes of the flagship config, with and
without TunableOp tuning, plus the quantize-pass overhead.

Run on the GPU box:
  PYTORCH_TUNABLEOP_ENABLED=1 python tools/fp8_bench.py
"""
import json
import os
import sys
import time

sys.path.insert(0, ".")
import torch

M = 32768  # tokens at batch 32, seq 1024, nmb 1
SHAPES = [
    ("qkv",   M, 2560, 7680),
    ("proj",  M, 2560, 2560),
    ("fc1",   M, 2560, 10240),
    ("fc2",   M, 10240, 2560),
    ("head",  M, 2560, 51200),
    # backward dX: [M,n] @ [n,k];  dW: [n,M] @ [M,k]
    ("qkv_dx",  M, 7680, 2560),
    ("fc1_dx",  M, 10240, 2560),
    ("head_dx", M, 51200, 2560),
    ("qkv_dw",  7680, M, 2560),
    ("fc1_dw",  10240, M, 2560),
    ("head_dw", 51200, M, 2560),
]


def bench(fn, warmup=5, rep=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(rep):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / rep


def main():
    dev = "cuda"
    results = {}
    for name, m, k, n in SHAPES:
        a = torch.randn(m, k, device=dev, dtype=torch.bfloat16)
        b = torch.randn(n, k, device=dev, dtype=torch.bfloat16)
        bt = b.t()  # (k, n) col-major view
        t_bf16 = bench(lambda: torch.matmul(a, bt))
        flops = 2.0 * m * k * n

        amax_a = a.abs().amax().float() / 448.0
        amax_b = b.abs().amax().float() / 448.0
        aq = (a.float() / amax_a).clamp(-448, 448).to(torch.float8_e4m3fn)
        bq = (b.float() / amax_b).clamp(-448, 448).to(torch.float8_e4m3fn)
        bqt = bq.t()
        t_fp8 = bench(lambda: torch._scaled_mm(
            aq, bqt, scale_a=amax_a, scale_b=amax_b,
            out_dtype=torch.bfloat16))
        # e5m2 activation-grad variant (backward recipe)
        aq5 = (a.float() / amax_a).clamp(-57344, 57344).to(
            torch.float8_e5m2)
        try:
            t_fp8_e5 = bench(lambda: torch._scaled_mm(
                aq5, bqt, scale_a=amax_a, scale_b=amax_b,
                out_dtype=torch.bfloat16))
        except Exception as e:
            t_fp8_e5 = None
        # quantize-pass cost (amax + cast, unfused baseline)
        t_q = bench(lambda: (a.float() / amax_a).clamp(-448, 448).to(
            torch.float8_e4m3fn))
        results[name] = {
            "bf16_ms": t_bf16 * 1e3, "bf16_tf": flops / t_bf16 / 1e12,
            "fp8_ms": t_fp8 * 1e3, "fp8_tf": flops / t_fp8 / 1e12,
            "fp8_e5m2_ms": t_fp8_e5 * 1e3 if t_fp8_e5 else None,
            "quant_ms": t_q * 1e3,
            "speedup": t_bf16 / t_fp8,
        }
        r = results[name]
        print(f"{name:8s} bf16 {r['bf16_tf']:7.0f} TF  "
              f"fp8 {r['fp8_tf']:7.0f} TF  x{r['speedup']:.2f}  "
              f"quant {r['quant_ms']:.2f} ms")
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/fp8_bench.json", "w") as f:
        json.dump(results, f, indent=2)
    tot_bf = sum(r["bf16_ms"] for r in results.values())
    tot_f8 = sum(r["fp8_ms"] for r in results.values())
    print(f"TOTAL bf16 {tot_bf:.1f} ms  fp8 {tot_f8:.1f} ms "
          f"(x{tot_bf/tot_f8:.2f})")


if __name__ == "__main__":
    main()
