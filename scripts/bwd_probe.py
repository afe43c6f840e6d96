import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math, time
import torch
import metis_amd._hip_ops as ext

B, H, S, D = 4, 32, 2048, 80
q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q); do = torch.randn_like(q)
sc = 1 / math.sqrt(D)
o, lse = ext.attn_fwd(q, k, v, sc)
delta = (do.float() * o.float()).sum(-1).contiguous()
def bench(fn, iters=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters
fl = 4 * B * H * S * S * D / 2 * 2.5
t = bench(lambda: ext.attn_bwd(q, k, v, do, lse, delta, sc))
print(f"attn bwd D80: {fl/t/1e12:.0f} TF/s-equiv")

B, H, S, D = 4, 32, 2048, 128
q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q); do = torch.randn_like(q)
sc = 1 / math.sqrt(D)
o, lse = ext.attn_fwd(q, k, v, sc)
delta = (do.float() * o.float()).sum(-1).contiguous()
fl = 4 * B * H * S * S * D / 2 * 2.5
t = bench(lambda: ext.attn_bwd(q, k, v, do, lse, delta, sc))
print(f"attn bwd D128: {fl/t/1e12:.0f} TF/s-equiv")
