"""Homogeneous-cluster planner entry point (reference: cost_homo_cluster.py).

The reference entry point crashes as shipped;
this is the fixed behavior: uniform plan sweep filtered to the requested
gbs, ranked by estimated iteration time. Optionally validates the cost
model against measured runtimes (--evaluation_data_path).
"""

from __future__ import annotations

import logging
from typing import List, Optional, Tuple

from metis_amd.cli.args import parse
from metis_amd.cluster import ClusterSpec
from metis_amd.config import ModelConfig, PlannerArgs
from metis_amd.planner.cost import HomoCostEstimator
from metis_amd.planner.plans import UniformPlan
from metis_amd.planner.uniform import uniform_plans
from metis_amd.planner.validate import CostValidator, plan_key
from metis_amd.planner.volume import make_volume
from metis_amd.profiles import ProfileStore

log = logging.getLogger(__name__)


def search_homo_cluster(
    cluster: ClusterSpec,
    profiles: ProfileStore,
    model_config: ModelConfig,
    planner_args: PlannerArgs,
    device_type: Optional[str] = None,
) -> List[Tuple[UniformPlan, float, bool]]:
    """Sweep uniform plans at gbs == planner_args.gbs; returns
    (plan, cost_ms, oom) tuples (unprofiled points skipped)."""
    volume = make_volume(
        model_config,
        profiles.model.parameters_per_layer_bytes,
        planner_args.activation_dtype_bytes,
    )
    estimator = HomoCostEstimator(profiles, model_config, volume, cluster, planner_args)
    dtype = device_type or profiles.device_type_names[0]

    results: List[Tuple[UniformPlan, float, bool]] = []
    for plan in uniform_plans(
        cluster.total_devices, planner_args.max_profiled_tp_degree, planner_args.gbs
    ):
        if plan.gbs != planner_args.gbs:
            continue
        try:
            cost, _stage_memory, oom = estimator.get_cost(plan, dtype)
        except KeyError as e:
            log.debug("skipping unprofiled plan %s: %s", plan, e)
            continue
        results.append((plan, cost, oom))
    return results


def main(argv: Optional[List[str]] = None) -> List[Tuple[UniformPlan, float, bool]]:
    args, model_config, planner_args = parse(argv)
    cluster = ClusterSpec(args.hostfile_path, args.clusterfile_path)
    profiles = ProfileStore.load_dir(args.profile_data_path)

    results = search_homo_cluster(cluster, profiles, model_config, planner_args)
    ranked = sorted(results, key=lambda r: r[1])
    if args.top_k:
        ranked = ranked[: args.top_k]

    print("rank, cost, plan")
    for idx, (plan, cost, oom) in enumerate(ranked):
        suffix = "  [OOM]" if oom else ""
        print(f"{idx + 1}, {cost}, {plan}{suffix}")

    if args.json_out:
        import json as _json

        with open(args.json_out, "w") as fh:
            _json.dump({"num_plans": len(results), "plans": [
                {"rank": i + 1, "cost_ms": cost, "oom": oom,
                 "plan": {"dp": p.dp, "tp": p.tp, "pp": p.pp,
                          "mbs": p.mbs, "gbs": p.gbs}}
                for i, (p, cost, oom) in enumerate(ranked)
            ]}, fh, indent=2)
        print(f"wrote {args.json_out}")

    if args.evaluation_data_path:
        validator = CostValidator(args.evaluation_data_path)
        estimates = {
            plan_key(p.dp, p.tp, p.pp, p.mbs, p.gbs): cost for p, cost, _ in results
        }
        v = validator.validate(estimates)
        print(
            f"cost-model validation: n={v.num_validated}, "
            f"mean_abs_error={v.mean_abs_error_pct:.2f}%, "
            f"max_abs_error={v.max_abs_error_pct:.2f}%, "
            f"within_tolerance={v.num_within_tolerance}/{v.num_validated}"
        )
    return ranked


if __name__ == "__main__":
    main()
