"""Pick the best (dp, tp, pp, mbs) plan for N MI355X GPUs from profiles.

Used by bench.py (--plan-search) and standalone: given the MI355X profile
directory and an N-GPU single-node clusterfile, run the homogeneous
uniform-plan sweep and return the cheapest plan that divides N.
"""

from __future__ import annotations

import json
import os
import tempfile
from typing import Optional, Tuple

from metis_amd.cluster import ClusterSpec
from metis_amd.config import ModelConfig, PlannerArgs
from metis_amd.cli.homo_cluster import search_homo_cluster
from metis_amd.profiles import ProfileStore


def single_node_cluster(
    n_gpus: int,
    device_type: str = "MI355X",
    intra_bandwidth: float = 130.0,
    inter_bandwidth: float = 40.0,
    memory_gb: float = 288.0,
) -> ClusterSpec:
    """An in-memory single-node clusterfile for N local GPUs."""
    tmp = tempfile.mkdtemp(prefix="metis_cluster_")
    host = os.path.join(tmp, "hostfile")
    with open(host, "w") as fh:
        fh.write(f"127.0.0.1 slots={n_gpus}\n")
    cf = os.path.join(tmp, "clusterfile.json")
    with open(cf, "w") as fh:
        json.dump({"127.0.0.1": {
            "instance_type": device_type,
            "intra_bandwidth": intra_bandwidth,
            "inter_bandwidth": inter_bandwidth,
            "memory": memory_gb,
        }}, fh)
    return ClusterSpec(host, cf)


def best_plan(
    profile_dir: str,
    model_config: ModelConfig,
    n_gpus: int,
    gbs: int,
    device_type: str = "MI355X",
    max_tp: int = 8,
    max_bs: int = 16,
    comm_bench_path: Optional[str] = None,
    intra_bandwidth: Optional[float] = None,
    alpha_us: Optional[float] = None,
) -> Optional[Tuple[int, int, int, int, float]]:
    """Returns (dp, tp, pp, mbs, est_ms) of the cheapest feasible plan, or
    None when no profiled plan fits. Explicit ``intra_bandwidth``/
    ``alpha_us`` (e.g. measured in-process by bench.py) override the
    comm_bench.json suggestion."""
    intra = 130.0
    alpha = 20.0
    if comm_bench_path and os.path.exists(comm_bench_path):
        with open(comm_bench_path) as fh:
            sugg = json.load(fh).get("clusterfile_suggestion", {})
        intra = sugg.get("intra_bandwidth", intra)
        alpha = sugg.get("alpha_us", alpha)
    if intra_bandwidth is not None:
        intra = intra_bandwidth
    if alpha_us is not None:
        alpha = alpha_us

    cluster = single_node_cluster(n_gpus, device_type, intra_bandwidth=intra)
    store = ProfileStore.load_dir(profile_dir, optimizer_scale=1.0)
    # MI355X flow: measured-accumulation microbatch model + bs
    # interpolation (falls back to parity on profiles without the keys)
    args = PlannerArgs(gbs=gbs, max_profiled_tp_degree=max_tp,
                       max_profiled_batch_size=max_bs,
                       comm_model="alpha_beta", alpha_us=alpha,
                       microbatch_model="marginal", interpolate_bs=True)
    results = search_homo_cluster(cluster, store, model_config, args,
                                  device_type=device_type)
    feasible = [(p, c) for p, c, oom in results if not oom]
    if not feasible:
        return None
    plan, cost = min(feasible, key=lambda r: r[1])
    return plan.dp, plan.tp, plan.pp, plan.mbs, cost


def estimate_plan(
    profile_dir: str,
    model_config: ModelConfig,
    n_gpus: int,
    gbs: int,
    *,
    dp: int,
    tp: int,
    pp: int,
    mbs: int,
    device_type: str = "MI355X",
    comm_bench_path: Optional[str] = None,
    intra_bandwidth: Optional[float] = None,
    alpha_us: Optional[float] = None,
    schedule: str = "gpipe",
    vpp: int = 1,
) -> Optional[float]:
    """Cost-model estimate (ms) for one specific plan; None if unprofiled."""
    from metis_amd.planner.cost import HomoCostEstimator
    from metis_amd.planner.plans import UniformPlan
    from metis_amd.planner.volume import make_volume

    intra, alpha = 130.0, 20.0
    if comm_bench_path and os.path.exists(comm_bench_path):
        with open(comm_bench_path) as fh:
            sugg = json.load(fh).get("clusterfile_suggestion", {})
        intra = sugg.get("intra_bandwidth", intra)
        alpha = sugg.get("alpha_us", alpha)
    if intra_bandwidth is not None:
        intra = intra_bandwidth
    if alpha_us is not None:
        alpha = alpha_us
    try:
        cluster = single_node_cluster(n_gpus, device_type, intra_bandwidth=intra)
        store = ProfileStore.load_dir(profile_dir, optimizer_scale=1.0)
        volume = make_volume(model_config, store.model.parameters_per_layer_bytes)
        est = HomoCostEstimator(
            store, model_config, volume, cluster,
            PlannerArgs(gbs=gbs, max_profiled_tp_degree=max(tp, 8),
                        max_profiled_batch_size=max(mbs, 16),
                        comm_model="alpha_beta", alpha_us=alpha,
                        microbatch_model="marginal", interpolate_bs=True,
                        schedule=schedule, vpp=vpp),
        )
        cost, _, _ = est.get_cost(UniformPlan(dp, pp, tp, mbs, gbs), device_type)
        return cost
    except (KeyError, FileNotFoundError):
        return None
