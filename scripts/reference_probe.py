#!/usr/bin/env python3
"""Dev-only probe: run pieces of the READ-ONLY reference checkout in an
isolated process and emit JSON, so parity tests can compare this
framework's behavior against the reference without importing it into the
test process (and without copying any reference code).

Usage: python3 reference_probe.py <mode> <json-args>
Modes:
  device_groups  {"cases": [[num_stages, num_gpus, variance, max_permute_len], ...]}
  uniform_plans  {"num_devices": N, "max_tp": T, "max_gbs": G, "limit": L}
  compute_balancer {"cases": [[num_stage, num_layer, capacities, demands], ...]}
  homo_costs     {"hostfile":..., "clusterfile":..., "profile_dir":..., "gbs":...,
                  "max_tp":..., "model": {...}}
"""

import json
import sys

REFERENCE = "/root/reference"
sys.path.insert(0, REFERENCE)


def device_groups(args):
    from search_space.device_group import gen_device_group_shapes, gen_dgroups_for_stages_with_variance

    out = []
    for num_stages, num_gpus, variance, max_permute_len in args["cases"]:
        shapes = gen_device_group_shapes(num_gpus)
        groups = gen_dgroups_for_stages_with_variance(
            num_stages=num_stages, num_gpus=num_gpus, group_shapes=shapes,
            variance=variance, max_permute_len=max_permute_len)
        out.append(sorted(map(tuple, groups)))
    return out


def uniform_plans(args):
    from search_space.plan import UniformPlanGenerator

    plans = []
    gen = UniformPlanGenerator(args["num_devices"], args["max_tp"], args["max_gbs"])
    for plan in gen:
        plans.append([plan.dp, plan.pp, plan.tp, plan.mbs, plan.gbs])
        if len(plans) >= args.get("limit", 100000):
            break
    return plans


def compute_balancer(args):
    from model.load_balancer import LayerComputeBalancer

    out = []
    for num_stage, num_layer, capacities, demands in args["cases"]:
        bal = LayerComputeBalancer(num_stage, num_layer, list(capacities), list(demands))
        partition, sc_demand = bal.run()
        out.append([partition, sc_demand])
    return out


def data_balancer(args):
    """Fuzz DataLoadBalancer: cases carry synthetic per-type layer-compute
    totals; build the minimal profile_data dict the reference reads."""
    from model.load_balancer import DataLoadBalancer

    out = []
    for device_types, dp_deg, tp_deg, bs, totals in args["cases"]:
        profile_data = {
            f"DeviceType.{t}": {
                f"tp{tp_deg}_bs1": {"time": {"layer-computes": [v]}}
            }
            for t, v in totals.items()
        }
        bal = DataLoadBalancer(profile_data, None)
        out.append(bal.partition_data(device_types, (dp_deg, tp_deg), bs))
    return out


def homo_costs(args):
    from gpu_cluster import GPUCluster
    from data_loader import ProfileDataLoader
    from model.cost_estimator import HomoCostEstimator
    from model.activation_parameter import GPTActivationAndParam
    from search_space.plan import UniformPlanGenerator
    from utils import ModelConfig

    cluster = GPUCluster(hostfile_path=args["hostfile"], clusterfile_path=args["clusterfile"])
    loader = ProfileDataLoader(args["profile_dir"])
    profile_data, device_types = loader.load_profile_data_all()
    m = args["model"]
    mc = ModelConfig(model_name=m["model_name"], num_layers=m["num_layers"],
                     sequence_length=m["sequence_length"], vocab_size=m["vocab_size"],
                     hidden_size=m["hidden_size"], attention_head_size=m["attention_head_size"])
    volume = GPTActivationAndParam(mc, profile_data["model"]["parameters"])
    est = HomoCostEstimator(profile_data, mc, volume, cluster)
    rows = []
    for plan in UniformPlanGenerator(cluster.get_total_num_devices(), args["max_tp"], args["gbs"]):
        if plan.gbs != args["gbs"]:
            continue
        try:
            cost, _mem, oom = est.get_cost(plan, device_types[0])
        except KeyError:
            continue
        rows.append([plan.dp, plan.pp, plan.tp, plan.mbs, cost, bool(oom)])
    return {"rows": rows, "model_file_order": loader.profile_data_list}


MODES = {
    "data_balancer": data_balancer,
    "device_groups": device_groups,
    "uniform_plans": uniform_plans,
    "compute_balancer": compute_balancer,
    "homo_costs": homo_costs,
}

if __name__ == "__main__":
    mode = sys.argv[1]
    args = json.loads(sys.argv[2]) if len(sys.argv) > 2 else {}
    import io, contextlib
    buf = io.StringIO()
    with contextlib.redirect_stdout(buf):  # reference prints a lot
        result = MODES[mode](args)
    sys.stdout.write(json.dumps(result))
