"""End-to-end cost-model validation on one MI355X (BASELINE config #2).

Estimates uniform-plan costs from the profiles in profiles/mi355x, runs
the same plans for real with the plan runner, and reports the cost-model
error % (the north-star metric) for THREE estimator modes side by side:

* parity    — reference formula (fb_sync charged per microbatch)
* marginal  — measured accumulation marginal + once-per-iteration
              residual (profile extension keys fwd_bwd_{1,2}mb_ms)
* marginal+interp — same, with bs interpolation so non-profiled mbs
              points (e.g. mbs=3 at gbs=12) are also validated

1-GPU plans only here; the multi-GPU sweep runs under torchrun via
metis_amd.cli.plan_runner.

Run (on a GPU box):  python scripts/validate_cost_model.py
Env: MODEL=gpt2-small GBS=8 STEPS=8 PROFILE_DIR=...
"""

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from metis_amd.cli.plan_search import single_node_cluster  # noqa: E402
from metis_amd.config import ModelConfig, PlannerArgs  # noqa: E402
from metis_amd.models.gpt import MODEL_SPECS  # noqa: E402
from metis_amd.planner.cost import HomoCostEstimator  # noqa: E402
from metis_amd.planner.plans import UniformPlan  # noqa: E402
from metis_amd.planner.validate import plan_key  # noqa: E402
from metis_amd.planner.volume import GPTVolume  # noqa: E402
from metis_amd.profiles import ProfileStore  # noqa: E402
from metis_amd.runtime.comm import init_parallel  # noqa: E402
from metis_amd.runtime.runner import PlanRunner  # noqa: E402

MODEL = os.environ.get("MODEL", "gpt2-small")
PROFILE_DIR = os.environ.get("PROFILE_DIR", f"profiles/mi355x/{MODEL}")
GBS = int(os.environ.get("GBS", "8"))
STEPS = int(os.environ.get("STEPS", "8"))
MAX_BS = int(os.environ.get("MAX_BS", "8"))

MODES = {
    "parity": {},
    "marginal": {"microbatch_model": "marginal"},
    "marginal_interp": {"microbatch_model": "marginal",
                        "interpolate_bs": True},
}


def main() -> None:
    spec = MODEL_SPECS[MODEL]
    mc = ModelConfig(spec.name, spec.profile_num_layers, spec.hidden_size,
                     spec.seq_length, spec.vocab_size)
    store = ProfileStore.load_dir(PROFILE_DIR, optimizer_scale=1.0)
    cluster = single_node_cluster(1)
    volume = GPTVolume(mc, store.model.parameters_per_layer_bytes)

    mbs_list = [m for m in range(1, GBS + 1) if GBS % m == 0 and m <= MAX_BS]
    estimates = {mode: {} for mode in MODES}
    for mode, extra in MODES.items():
        est = HomoCostEstimator(
            store, mc, volume, cluster,
            PlannerArgs(gbs=GBS, max_profiled_tp_degree=1,
                        max_profiled_batch_size=MAX_BS, **extra))
        for mbs in mbs_list:
            try:
                cost, _, _ = est.get_cost(UniformPlan(1, 1, 1, mbs, GBS),
                                          "MI355X")
            except KeyError:
                continue
            estimates[mode][plan_key(1, 1, 1, mbs, GBS)] = cost

    ctx = init_parallel(dp=1, tp=1, pp=1)
    measured = {}
    for mbs in mbs_list:
        key = plan_key(1, 1, 1, mbs, GBS)
        if not any(key in estimates[m] for m in MODES):
            continue
        runner = PlanRunner(spec, ctx, mbs=mbs, gbs=GBS)
        ms = runner.timed_steps(STEPS, 3)
        measured[key] = ms
        row = " ".join(
            f"{m}:{estimates[m][key]:.2f}ms({(estimates[m][key]-ms)/ms*100:+.1f}%)"
            for m in MODES if key in estimates[m])
        print(f"{key}: measured {ms:.2f} ms | {row}")
        del runner
        torch.cuda.empty_cache()

    summary = {"model": MODEL, "gbs": GBS, "modes": {}}
    for mode in MODES:
        errs = {k: abs(estimates[mode][k] - v) / v * 100
                for k, v in measured.items() if k in estimates[mode]}
        if not errs:
            continue
        summary["modes"][mode] = {
            "mean_abs_error_pct": sum(errs.values()) / len(errs),
            "max_abs_error_pct": max(errs.values()),
            "num_validated": len(errs),
            "per_plan": {k: {"est_ms": estimates[mode][k],
                             "measured_ms": measured[k],
                             "err_pct": (estimates[mode][k] - measured[k])
                             / measured[k] * 100} for k in errs},
        }
    os.makedirs("gpurun_out", exist_ok=True)
    out = f"gpurun_out/cost_model_validation_{MODEL}_gbs{GBS}.json"
    with open(out, "w") as fh:
        json.dump(summary, fh, indent=2)
    print(json.dumps({m: {k: v for k, v in d.items() if k != "per_plan"}
                      for m, d in summary["modes"].items()}))


if __name__ == "__main__":
    main()
