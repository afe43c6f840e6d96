"""Decode-path microbench: tokens/s for KV-cache generation on 1 GPU.

A/B driver for METIS_DECODE_KERNEL (single-query decode attention
kernel vs the SDPA fallback): run once with the env unset and once =1.

Usage: python scripts/decode_bench.py [--model llama3-1b] [--new 64]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from metis_amd.cli.plan_runner import MODEL_SPECS
from metis_amd.models.gpt import GPTModel, GPTModelSpec
from metis_amd.models.llama import LlamaModel, LlamaModelSpec
from metis_amd.runtime.generate import generate


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-1b")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--prompt", type=int, default=512)
    ap.add_argument("--new", type=int, default=64)
    args = ap.parse_args()

    spec = MODEL_SPECS[args.model]
    cls = LlamaModel if isinstance(spec, LlamaModelSpec) else GPTModel
    torch.manual_seed(0)
    model = cls(spec, tp=1, dtype=torch.bfloat16).to("cuda")
    tokens = torch.randint(0, spec.vocab_size, (args.batch, args.prompt),
                           device="cuda")
    # warmup
    generate(model, tokens, 8, temperature=0.0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = generate(model, tokens, args.new, temperature=0.0)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    tps = args.batch * args.new / dt
    flag = os.environ.get("METIS_DECODE_KERNEL", "1(default)")
    print(f"decode {args.model} b{args.batch} p{args.prompt} n{args.new} "
          f"METIS_DECODE_KERNEL={flag}: {dt*1e3:.1f} ms, {tps:.0f} tok/s")
    assert out.shape == (args.batch, args.prompt + args.new)


if __name__ == "__main__":
    main()
