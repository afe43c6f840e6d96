"""Dev harness: localize flash-attention fwd errors (runs on GPU box)."""
import math
import sys

import torch

sys.path.insert(0, "/root/repo")
import metis_amd._hip_ops as ext


def check(b, h, hkv, s, d, mode="rand", tag=""):
    torch.manual_seed(0)
    q = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16)
    if mode == "same":
        k = q.clone()
        v = q.clone()
    else:
        k = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(b, hkv, s, d, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(d)
    o, lse = ext.attn_fwd(q, k, v, scale)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.float(), k.float(), v.float(), is_causal=True, scale=scale,
        enable_gqa=h != hkv)
    err = (o.float() - ref).abs()
    m = err.max().item()
    print(f"[{tag}] b{b} h{h} hkv{hkv} s{s} d{d} {mode}: max_err={m:.4f}")
    if m > 3e-2:
        # where: per q-row max error, bucketed by row % 64 (wave strips)
        per_row = err.amax(dim=(0, 1, 3))   # [S]
        bad_rows = (per_row > 3e-2).nonzero().flatten()
        print("   bad rows:", bad_rows[:12].tolist(), "…", len(bad_rows), "total")
        if len(bad_rows):
            r = int(bad_rows[0])
            per_d = err[:, :, r, :].amax(dim=(0, 1))
            bad_d = (per_d > 3e-2).nonzero().flatten()
            print(f"   row {r}: bad d cols:", bad_d[:16].tolist(), len(bad_d), "total")
            print("   row%64:", r % 64, "wave:", (r % 64) // 16, "qtile:", r // 64)
        # lse check
        scores = torch.einsum("bhsd,bhtd->bhst", q.float(),
                              k.float().repeat_interleave(h // hkv, 1) if h != hkv else k.float()) * scale
        mask = torch.tril(torch.ones(s, s, device="cuda", dtype=torch.bool))
        scores = scores.masked_fill(~mask, float("-inf"))
        ref_lse = torch.logsumexp(scores, -1)
        print("   lse max err:", (lse - ref_lse).abs().max().item())


check(2, 4, 4, 256, 64, "rand", "pass-before")
check(2, 4, 4, 512, 64, "rand", "S512")
check(1, 8, 2, 512, 128, "rand", "gqa128")
check(2, 4, 4, 256, 80, "rand", "d80")
check(1, 4, 4, 256, 64, "same", "kvq-same")
check(1, 1, 1, 128, 64, "rand", "small")
check(1, 1, 1, 256, 128, "rand", "d128-small")
check(1, 1, 1, 256, 80, "rand", "d80-small")
