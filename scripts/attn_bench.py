"""Attention fwd/bwd microbench: our HIP flash kernels vs PyTorch SDPA.

Prints one line per (D, direction, impl) with ms and achieved TF/s
(causal FLOP count: 2*B*H*S^2*D per matmul pair, halved for causal,
x2 for QK^T+PV; bwd counts 2.5x fwd as is conventional).

Usage: python scripts/attn_bench.py [--iters 20] [--json OUT]
"""
import argparse
import json
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

import metis_amd._hip_ops as ext


def bench(fn, iters):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def run(B, H, S, D, iters, results):
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    do = torch.randn_like(q)
    sc = 1 / math.sqrt(D)
    fwd_fl = 4 * B * H * S * S * D / 2            # causal
    bwd_fl = fwd_fl * 2.5

    t = bench(lambda: ext.attn_fwd(q, k, v, sc), iters)
    results.append(dict(op="fwd", impl="hip", B=B, H=H, S=S, D=D,
                        ms=t * 1e3, tflops=fwd_fl / t / 1e12))

    t = bench(lambda: F.scaled_dot_product_attention(
        q, k, v, is_causal=True), iters)
    results.append(dict(op="fwd", impl="sdpa", B=B, H=H, S=S, D=D,
                        ms=t * 1e3, tflops=fwd_fl / t / 1e12))

    o, lse = ext.attn_fwd(q, k, v, sc)
    delta = (do.float() * o.float()).sum(-1).contiguous()
    t = bench(lambda: ext.attn_bwd(q, k, v, do, lse, delta, sc), iters)
    results.append(dict(op="bwd", impl="hip", B=B, H=H, S=S, D=D,
                        ms=t * 1e3, tflops=bwd_fl / t / 1e12))

    qs = q.clone().requires_grad_(True)
    ks = k.clone().requires_grad_(True)
    vs = v.clone().requires_grad_(True)
    out = F.scaled_dot_product_attention(qs, ks, vs, is_causal=True)

    def sdpa_bwd():
        qs.grad = ks.grad = vs.grad = None
        torch.autograd.grad(out, (qs, ks, vs), do, retain_graph=True)

    t = bench(sdpa_bwd, iters)
    results.append(dict(op="bwd", impl="sdpa", B=B, H=H, S=S, D=D,
                        ms=t * 1e3, tflops=bwd_fl / t / 1e12))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--json", type=str, default=None)
    args = ap.parse_args()
    results = []
    # flagship-relevant shapes: gpt3-2.7b (H=32, D=80), llama/gpt D=128, D=64
    for B, H, S, D in [(4, 32, 2048, 64), (4, 32, 2048, 80),
                       (4, 32, 2048, 128), (2, 32, 4096, 128)]:
        run(B, H, S, D, args.iters, results)
    for r in results:
        print(f"{r['op']:>3} {r['impl']:>4} B{r['B']} H{r['H']} S{r['S']} "
              f"D{r['D']:<3} {r['ms']:8.3f} ms  {r['tflops']:7.1f} TF/s")
    if args.json:
        with open(args.json, "w") as f:
            json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
