"""Run concrete (dp, tp, pp) plans and record measured iteration times.

Closes the loop the reference left open (its EstimateCostValidator had no
data path, quirk Q3): execute plans on real GPUs with synthetic data and
emit the measured-runs JSON that metis_amd.planner.validate consumes.

Single GPU:  python -m metis_amd.cli.plan_runner --model gpt2-small \
                 --plans "1,1,1,1,4;1,1,1,2,4" --out measured.json
Multi GPU:   torchrun --nproc-per-node N -m metis_amd.cli.plan_runner ...
             (every plan must satisfy dp*tp*pp == N)
"""

from __future__ import annotations

import argparse
import json
import os
from typing import List, Tuple

import torch
import torch.distributed as dist

from metis_amd.models.gpt import MODEL_SPECS as _GPT_SPECS
from metis_amd.models.llama import LLAMA_SPECS  # noqa: E402
from metis_amd.models.moe import MOE_SPECS  # noqa: E402

MODEL_SPECS = {**_GPT_SPECS, **LLAMA_SPECS, **MOE_SPECS}
from metis_amd.planner.validate import plan_key
from metis_amd.runtime.comm import init_parallel
from metis_amd.runtime.runner import PlanRunner


def parse_plans(text: str) -> List[Tuple[int, int, int, int, int]]:
    """"dp,tp,pp,mbs,gbs;..." -> [(dp, tp, pp, mbs, gbs), ...]."""
    plans = []
    for chunk in text.split(";"):
        chunk = chunk.strip()
        if not chunk:
            continue
        dp, tp, pp, mbs, gbs = (int(x) for x in chunk.split(","))
        plans.append((dp, tp, pp, mbs, gbs))
    return plans


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="gpt2-small", choices=sorted(MODEL_SPECS))
    p.add_argument("--plans", required=True,
                   help='semicolon-separated "dp,tp,pp,mbs,gbs" tuples')
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--out", default="profiles/measured_runs.json")
    p.add_argument("--schedule", default="gpipe", choices=("gpipe", "1f1b", "interleaved"))
    p.add_argument("--vpp", type=int, default=2,
                   help="virtual chunks per rank for --schedule interleaved")
    p.add_argument("--recompute", action="store_true",
                   help="per-block activation recomputation")
    p.add_argument("--zero1", action="store_true",
                   help="shard optimizer state over the DP group")
    p.add_argument("--sp", action="store_true",
                   help="sequence parallelism in the TP norm regions (GPT)")
    p.add_argument("--layer-partition", default=None,
                   help='non-uniform stage boundaries from the hetero '
                        'planner, e.g. "0,13,34" (cumulative, profile '
                        'layer convention); default: uniform split')
    args = p.parse_args()
    layer_partition = ([int(x) for x in args.layer_partition.split(",")]
                       if args.layer_partition else None)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    spec = MODEL_SPECS[args.model]
    runs = []
    for dp, tp, pp, mbs, gbs in parse_plans(args.plans):
        if dp * tp * pp != world:
            print(f"skip dp{dp}_tp{tp}_pp{pp}: needs world {dp * tp * pp}, have {world}")
            continue
        ctx = init_parallel(dp=dp, tp=tp, pp=pp)
        if layer_partition is not None:
            assert len(layer_partition) == pp + 1, (
                "--layer-partition needs pp+1 boundaries")
        runner = PlanRunner(spec, ctx, mbs=mbs, gbs=gbs,
                            layer_partition=layer_partition,
                            schedule=args.schedule,
                            recompute=args.recompute,
                            zero1=args.zero1, sp=args.sp, vpp=args.vpp)
        ms = runner.timed_steps(args.steps, args.warmup)
        if dist.is_initialized():
            t = torch.tensor([ms], dtype=torch.float64,
                             device=ctx.device if ctx.device else "cpu")
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            ms = float(t.item())
        if ctx.rank == 0:
            print(f"{plan_key(dp, tp, pp, mbs, gbs)}: {ms:.2f} ms/step")
            runs.append({
                "plan": {"dp": dp, "tp": tp, "pp": pp, "mbs": mbs, "gbs": gbs},
                "measured_ms": ms,
                "model": args.model,
            })
        del runner
        if torch.cuda.is_available():
            torch.cuda.empty_cache()

    rank = int(os.environ.get("RANK", "0"))
    if rank == 0 and runs:
        os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
        doc = {"runs": runs}
        if os.path.exists(args.out):
            try:
                with open(args.out) as fh:
                    doc["runs"] = json.load(fh)["runs"] + runs
            except Exception:
                pass
        with open(args.out, "w") as fh:
            json.dump(doc, fh, indent=2)
        print(f"wrote {args.out} ({len(doc['runs'])} runs)")

    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
