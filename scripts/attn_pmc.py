"""Minimal attention kernel exerciser for rocprofv3 PMC runs: a few
iterations of fwd + bwd at the flagship shapes, nothing else."""
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import metis_amd._hip_ops as ext

for B, H, S, D in [(4, 32, 2048, 80), (4, 32, 2048, 128)]:
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    do = torch.randn_like(q)
    sc = 1 / math.sqrt(D)
    o, lse = ext.attn_fwd(q, k, v, sc)
    delta = (do.float() * o.float()).sum(-1).contiguous()
    for _ in range(3):
        ext.attn_fwd(q, k, v, sc)
        ext.attn_bwd(q, k, v, do, lse, delta, sc)
    torch.cuda.synchronize()
print("PMC_RUN_OK")
