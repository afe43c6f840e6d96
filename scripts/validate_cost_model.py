"""End-to-end cost-model validation on one MI355X (BASELINE config #2).

Estimates uniform-plan costs from the profiles in profiles/mi355x, runs
the same plans for real with the plan runner, and reports the cost-model
error % (the north-star metric). 1-GPU plans only here; the multi-GPU
sweep runs under torchrun via metis_amd.cli.plan_runner.

Run (on a GPU box):  python scripts/validate_cost_model.py
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
from metis_amd.planner.validate import CostValidator, plan_key  # noqa: E402
from metis_amd.planner.volume import GPTVolume  # noqa: E402
from metis_amd.profiles import ProfileStore  # noqa: E402
from metis_amd.runtime.comm import init_parallel  # noqa: E402
from metis_amd.runtime.runner import PlanRunner  # noqa: E402

MODEL = os.environ.get("MODEL", "gpt2-small")
PROFILE_DIR = os.environ.get("PROFILE_DIR", f"profiles/mi355x/{MODEL}")
GBS = int(os.environ.get("GBS", "8"))
STEPS = int(os.environ.get("STEPS", "8"))


def main() -> None:
    spec = MODEL_SPECS[MODEL]
    mc = ModelConfig(spec.name, spec.profile_num_layers, spec.hidden_size,
                     spec.seq_length, spec.vocab_size)
    store = ProfileStore.load_dir(PROFILE_DIR, optimizer_scale=1.0)
    cluster = single_node_cluster(1)
    volume = GPTVolume(mc, store.model.parameters_per_layer_bytes)
    est = HomoCostEstimator(store, mc, volume, cluster,
                            PlannerArgs(gbs=GBS, max_profiled_tp_degree=1,
                                        max_profiled_batch_size=8))

    # estimate all profiled 1-GPU plans at this gbs
    estimates = {}
    for mbs in (1, 2, 4, 8):
        if GBS % mbs:
            continue
        try:
            cost, _, _ = est.get_cost(UniformPlan(1, 1, 1, mbs, GBS), "MI355X")
        except KeyError:
            continue
        estimates[plan_key(1, 1, 1, mbs, GBS)] = cost

    # measure the same plans
    ctx = init_parallel(dp=1, tp=1, pp=1)
    validator = CostValidator(error_threshold_pct=15.0)
    for mbs in (1, 2, 4, 8):
        key = plan_key(1, 1, 1, mbs, GBS)
        if key not in estimates:
            continue
        runner = PlanRunner(spec, ctx, mbs=mbs, gbs=GBS)
        ms = runner.timed_steps(STEPS, 3)
        validator.add_measurement(key, ms)
        print(f"{key}: est {estimates[key]:.2f} ms, measured {ms:.2f} ms, "
              f"err {abs(estimates[key] - ms) / ms * 100:.1f}%")
        del runner
        torch.cuda.empty_cache()

    result = validator.validate(estimates)
    summary = {
        "model": MODEL,
        "gbs": GBS,
        "mean_abs_error_pct": result.mean_abs_error_pct,
        "max_abs_error_pct": result.max_abs_error_pct,
        "num_validated": result.num_validated,
        "per_plan": {k: {"est_ms": v[0], "measured_ms": v[1], "err_pct": v[2]}
                     for k, v in result.per_plan.items()},
    }
    os.makedirs("gpurun_out", exist_ok=True)
    out = f"gpurun_out/cost_model_validation_{MODEL}.json"
    with open(out, "w") as fh:
        json.dump(summary, fh, indent=2)
    print(json.dumps({k: v for k, v in summary.items() if k != "per_plan"}))


if __name__ == "__main__":
    main()
