"""Heterogeneous-cluster planner entry point (reference: cost_het_cluster.py).

Enumerates inter-stage x intra-stage plans, costs each with the hetero
estimator, prints a ranked table (output contract: SURVEY.md Appendix B)::

    rank, cost, node_sequence, device_groups, strategies(dp_deg, tp_deg),
    batches(number of batch), layer_partition
"""

from __future__ import annotations

import logging
from typing import List, Optional, Tuple

from metis_amd.cli.args import parse
from metis_amd.cluster import ClusterSpec
from metis_amd.config import ModelConfig, PlannerArgs
from metis_amd.planner.balancer import LayerLoadBalancer, StagePerformance
from metis_amd.planner.cost import HeteroCostEstimator
from metis_amd.planner.inter_stage import inter_stage_plans
from metis_amd.planner.intra_stage import intra_stage_plans
from metis_amd.planner.volume import GPTVolume
from metis_amd.profiles import ProfileStore

log = logging.getLogger(__name__)

PlanResult = Tuple[tuple, List[int], List[Tuple[int, int]], int, List[int], int, float]


def search_het_cluster(
    cluster: ClusterSpec,
    profiles: ProfileStore,
    model_config: ModelConfig,
    planner_args: PlannerArgs,
) -> List[PlanResult]:
    """Full heterogeneous plan search; returns unsorted
    (node_sequence, device_groups, strategies, batches, layer_partition,
    num_repartition, cost) tuples."""
    volume = GPTVolume(
        model_config,
        profiles.model.parameters_per_layer_bytes,
        planner_args.activation_dtype_bytes,
    )
    estimator = HeteroCostEstimator(profiles, model_config, volume, cluster, planner_args)
    layer_balancer = LayerLoadBalancer(cluster, profiles, model_config, planner_args.gbs)

    results: List[PlanResult] = []
    for inter_plan in inter_stage_plans(
        device_types=cluster.unique_device_types(),
        num_devices=cluster.total_devices,
        gbs=planner_args.gbs,
        num_layers=model_config.num_layers,
        variance=planner_args.min_group_scale_variance,
        max_permute_len=planner_args.max_permute_len,
    ):
        stage_perf = StagePerformance(model_config, profiles, cluster, inter_plan)
        rank_device_map = stage_perf.rank_device_map
        try:
            for intra_plan in intra_stage_plans(
                inter_plan,
                stage_perf,
                layer_balancer,
                planner_args.max_profiled_tp_degree,
                planner_args.max_profiled_batch_size,
            ):
                try:
                    cost = estimator.get_cost(
                        inter_plan, intra_plan.strategies, intra_plan.layer_partition,
                        rank_device_map,
                    )
                except KeyError as e:
                    log.debug("skipping unprofiled plan: %s", e)
                    continue
                if (planner_args.drop_incomplete_partitions
                        and intra_plan.layer_partition[-1] != model_config.num_layers):
                    # reference balancer quirk: a layer fell out of the
                    # partition; under-costed and unrunnable (config.py)
                    log.debug("dropping incomplete partition %s",
                              intra_plan.layer_partition)
                    continue
                results.append(
                    (
                        tuple(inter_plan.node_sequence),
                        list(inter_plan.device_groups),
                        list(intra_plan.strategies),
                        inter_plan.batches,
                        list(intra_plan.layer_partition),
                        intra_plan.num_repartition,
                        cost,
                    )
                )
        except KeyError as e:
            # unprofiled (tp, bs) hit inside stage-performance / balancing
            log.debug("skipping inter-stage plan: %s", e)
            continue
    return results


def main(argv: Optional[List[str]] = None) -> List[PlanResult]:
    args, model_config, planner_args = parse(argv)
    cluster = ClusterSpec(args.hostfile_path, args.clusterfile_path)
    profiles = ProfileStore.load_dir(args.profile_data_path)

    results = search_het_cluster(cluster, profiles, model_config, planner_args)

    ranked = sorted(results, key=lambda r: r[6])
    if args.top_k:
        ranked = ranked[: args.top_k]
    print(f"len(costs): {len(results)}")
    print(
        "rank, cost, node_sequence, device_groups, strategies(dp_deg, tp_deg), "
        "batches(number of batch), layer_partition"
    )
    for idx, r in enumerate(ranked):
        node_seq = [str(s) for s in r[0]]
        print(f"{idx + 1}, {r[6]}, {node_seq}, {r[1]}, {r[2]}, {r[3]}, {r[4]}")
    if args.json_out:
        import json as _json

        with open(args.json_out, "w") as fh:
            _json.dump({"num_plans": len(results), "plans": [
                {"rank": i + 1, "cost_ms": r[6],
                 "node_sequence": [str(s) for s in r[0]],
                 "device_groups": r[1],
                 "strategies": [list(st) for st in r[2]],
                 "batches": r[3], "layer_partition": r[4],
                 "num_repartition": r[5]}
                for i, r in enumerate(ranked)
            ]}, fh, indent=2)
        print(f"wrote {args.json_out}")
    return ranked


if __name__ == "__main__":
    main()
