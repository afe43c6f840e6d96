#!/usr/bin/env python3
"""Flagship benchmark: best-plan training iteration time, GPT-3 2.7B.

Contract (driver): `python bench.py --gpus N --steps K --warmup W` runs the
flagship training step on N GPUs of one node (launched via
torch.distributed.run for N > 1, one rank per GPU over RCCL). Rank 0
prints ONE JSON line with the measured iteration time.

Metric (BASELINE.json): best-plan measured iteration time (ms) for
GPT-3 2.7B on synthetic data / random-init weights, bf16. Weak scaling:
per-GPU batch is fixed, so gbs grows with N.
"""

from __future__ import annotations

import argparse
import json
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from metis_amd.models.gpt import MODEL_SPECS as _GPT_SPECS  # noqa: E402
from metis_amd.models.llama import LLAMA_SPECS  # noqa: E402
from metis_amd.models.moe import MOE_SPECS  # noqa: E402

MODEL_SPECS = {**_GPT_SPECS, **LLAMA_SPECS, **MOE_SPECS}
from metis_amd.runtime.comm import init_parallel  # noqa: E402
from metis_amd.runtime.runner import PlanRunner  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--model", default="gpt3-2.7b", choices=sorted(MODEL_SPECS))
    p.add_argument("--per-gpu-batch", type=int, default=16,
                   help="sequences per GPU per step (weak scaling)")
    p.add_argument("--mbs", type=int, default=16, help="microbatch size")
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--pp", type=int, default=1)
    p.add_argument("--plan-search", action="store_true",
                   help="pick (dp, tp, pp, mbs) with the planner from "
                        "--profile-dir instead of the flags")
    p.add_argument("--profile-dir", default=None,
                   help="defaults to profiles/mi355x/<model>")
    p.add_argument("--schedule", default="gpipe", choices=("gpipe", "1f1b", "interleaved"))
    p.add_argument("--vpp", type=int, default=2,
                   help="virtual chunks per rank for --schedule interleaved")
    p.add_argument("--recompute", action="store_true")
    p.add_argument("--zero1", action="store_true",
                   help="shard optimizer state over the DP group")
    p.add_argument("--sp", action="store_true",
                   help="sequence parallelism in the TP norm regions (GPT)")
    return p.parse_args()


def main() -> None:
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    n_gpus = max(world, args.gpus if world == 1 else world)
    spec = MODEL_SPECS[args.model]

    dp = n_gpus // (args.tp * args.pp)
    assert dp * args.tp * args.pp == n_gpus, "tp*pp must divide --gpus"
    gbs = args.per_gpu_batch * n_gpus // (args.tp * args.pp)
    mbs = min(args.mbs, max(gbs // dp, 1))
    tp, pp = args.tp, args.pp

    profile_dir = args.profile_dir or f"profiles/mi355x/{args.model}"
    est_ms = None
    mc = None
    if os.path.isdir(profile_dir):
        from metis_amd.config import ModelConfig
        from metis_amd.cli.plan_search import best_plan, estimate_plan

        moe = {}
        if hasattr(spec, "num_experts"):
            moe = dict(num_experts=spec.num_experts,
                       expert_weight_mul=(
                           3 if spec.expert_activation == "swiglu" else 2),
                       ffn_hidden_size=spec.ffn)
        mc = ModelConfig(
            model_name=spec.name,
            num_layers=spec.profile_num_layers,
            hidden_size=spec.hidden_size,
            sequence_length=spec.seq_length,
            vocab_size=spec.vocab_size,
            **moe,
        )
        if args.plan_search:
            found = best_plan(profile_dir, mc, n_gpus, gbs,
                              comm_bench_path="profiles/comm_bench.json")
            if found:
                dp, tp, pp, mbs, est_ms = found

    ctx = init_parallel(dp=dp, tp=tp, pp=pp)

    # At N > 1, calibrate the xGMI comm constants IN-PROCESS before the
    # timed region (a short rccl-tests-style all-reduce sweep over the
    # world group) and feed the measured bus bandwidth + latency into the
    # cost-model estimate — the planner's comm terms are then measured,
    # not clusterfile guesses. Reported in config for the record.
    comm_busbw = comm_alpha = None
    if n_gpus > 1 and torch.cuda.is_available() and dist.is_initialized():
        from metis_amd.profiler.comm_bench import bench_allreduce

        rows = bench_allreduce([1 << 16, 1 << 22, 1 << 26, 1 << 28],
                               iters=8, warmup=3)
        comm_busbw = round(max(r["busbw_GBps"] for r in rows), 1)
        comm_alpha = round(min(r["time_us"] for r in rows), 1)

    if mc is not None and est_ms is None:
        from metis_amd.cli.plan_search import estimate_plan

        # report the cost-model estimate for the plan we are running
        est_ms = estimate_plan(profile_dir, mc, n_gpus, gbs,
                               dp=dp, tp=tp, pp=pp, mbs=mbs,
                               comm_bench_path="profiles/comm_bench.json",
                               intra_bandwidth=comm_busbw,
                               alpha_us=comm_alpha,
                               schedule=args.schedule, vpp=args.vpp)
    runner = PlanRunner(spec, ctx, mbs=mbs, gbs=gbs,
                        schedule=args.schedule, recompute=args.recompute,
                        zero1=args.zero1, sp=args.sp, vpp=args.vpp)

    ms = runner.timed_steps(args.steps, args.warmup)

    # per-rank max: the slowest rank defines the iteration time
    if dist.is_initialized():
        t = torch.tensor([ms], dtype=torch.float64,
                         device=ctx.device if ctx.device else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        ms = float(t.item())

    if ctx.rank == 0:
        tokens_per_step = gbs * spec.seq_length
        result = {
            "metric": "best_plan_iter_time_ms",
            "value": ms,
            "unit": "ms",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms,
            "higher_is_better": False,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": gbs,
                "seq_len": spec.seq_length,
                "parallelism": f"dp{dp}_tp{tp}_pp{pp}",
                "microbatch": mbs,
                "tokens_per_s": tokens_per_step / (ms / 1000.0),
                "planner_estimate_ms": est_ms,
                "cost_model_error_pct": (
                    abs(est_ms - ms) / ms * 100.0 if est_ms else None
                ),
                "comm_busbw_GBps": comm_busbw,
                "comm_alpha_us": comm_alpha,
            },
        }
        print(json.dumps(result))

    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
