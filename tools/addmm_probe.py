import sys, time, torch
sys.path.insert(0, "/root/repo")
def t(fn, it=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/it*1e6
for (m,k,n) in [(32768,2560,7680),(32768,2560,2560),(32768,10240,2560)]:
    x = torch.randn(m,k, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(n,k, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    t1 = t(lambda: torch.matmul(x, w.t()) + b)
    t2 = t(lambda: torch.addmm(b, x, w.t()))
    t3 = t(lambda: torch.matmul(x, w.t()))
    print(f"[{m}x{k}x{n}] matmul+add {t1:8.1f}us  addmm {t2:8.1f}us  "
          f"matmul-only {t3:8.1f}us")
