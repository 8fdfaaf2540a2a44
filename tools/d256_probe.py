import sys, time, math, torch
sys.path.insert(0, "/root/repo")
from alpa_amd.ops._backend import hip_ops
ext = hip_ops()
def t(fn, it=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/it
for (B,H,S,D) in [(8,16,512,256),(8,16,2048,256),(1,16,2048,256)]:
    q = torch.randn(B,H,S,D, device="cuda", dtype=torch.bfloat16)*0.3
    k = torch.randn_like(q)*0.3; v = torch.randn_like(q)
    sc = 1.0/math.sqrt(D)
    o1, l1 = ext.attn_fwd(q,k,v,True,sc)
    o2, l2 = ext.attn_fwd_blocked(q,k,v,True,sc)
    err = (o1.float()-o2.float()).abs().max().item()
    t1 = t(lambda: ext.attn_fwd(q,k,v,True,sc))
    t2 = t(lambda: ext.attn_fwd_blocked(q,k,v,True,sc))
    fl = 4*B*H*S*S*D*0.5
    print(f"B{B} S{S}: fused {t1*1e6:8.1f}us {fl/t1/1e12:6.1f}TF  "
          f"blocked {t2*1e6:8.1f}us {fl/t2/1e12:6.1f}TF  maxdiff {err:.4f}")
# decode shape: q len 1, kv 2048
B,H,D = 16,16,256
kc = torch.randn(B,H,2048,D, device="cuda", dtype=torch.bfloat16)*0.3
vc = torch.randn_like(kc)
q1 = torch.randn(B,H,1,D, device="cuda", dtype=torch.bfloat16)*0.3
sc = 1.0/math.sqrt(D)
o1,_ = ext.attn_fwd(q1,kc,vc,False,sc)
o2,_ = ext.attn_fwd_blocked(q1,kc,vc,False,sc)
print("decode maxdiff", (o1.float()-o2.float()).abs().max().item(),
      "fused", t(lambda: ext.attn_fwd(q1,kc,vc,False,sc))*1e6, "us",
      "blocked", t(lambda: ext.attn_fwd_blocked(q1,kc,vc,False,sc))*1e6, "us")
