"""Bit-parity tests against the read-only reference checkout.

These run only where /root/reference exists (the dev container; the
driver's CPU test run). They execute the reference in a SUBPROCESS via
scripts/reference_probe.py — no reference code is imported into this
process or copied into the repo.
"""

import json
import os
import random
import subprocess
import sys

import pytest

pytestmark = pytest.mark.reference

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
PROBE = os.path.join(REPO, "scripts", "reference_probe.py")
SAMPLES = "/root/reference/profile_data_samples"


def probe(mode, args):
    out = subprocess.run(
        [sys.executable, PROBE, mode, json.dumps(args)],
        capture_output=True, text=True, check=True,
    )
    return json.loads(out.stdout)


def test_device_groups_parity():
    from metis_amd.planner.groups import power_of_two_shapes, stage_device_groups

    cases = []
    for num_gpus in (4, 8, 16, 32):
        for num_stages in (1, 2, 3, 4, 5):
            for variance in (0.5, 1):
                for mpl in (2, 4, 6):
                    cases.append([num_stages, num_gpus, variance, mpl])
    ref = probe("device_groups", {"cases": cases})
    for case, ref_groups in zip(cases, ref):
        num_stages, num_gpus, variance, mpl = case
        mine = stage_device_groups(
            num_stages, num_gpus, power_of_two_shapes(num_gpus), variance, mpl
        )
        assert sorted(map(tuple, mine)) == sorted(map(tuple, ref_groups)), case


def test_uniform_plans_parity():
    from metis_amd.planner.uniform import uniform_plans

    for n, max_tp, max_gbs in ((8, 4, 16), (16, 8, 64), (4, 2, 8), (2, 4, 4)):
        ref = probe("uniform_plans", {"num_devices": n, "max_tp": max_tp,
                                      "max_gbs": max_gbs})
        mine = [[p.dp, p.pp, p.tp, p.mbs, p.gbs] for p in uniform_plans(n, max_tp, max_gbs)]
        assert mine == ref, (n, max_tp, max_gbs)


def test_compute_balancer_parity():
    from metis_amd.planner.balancer import LayerComputeBalancer

    rng = random.Random(1234)
    cases = []
    for _ in range(40):
        num_stage = rng.randint(1, 6)
        num_layer = rng.choice([8, 10, 12, 24])
        raw = [rng.uniform(0.3, 1.5) for _ in range(num_stage)]
        total = sum(raw)
        caps = [r / total for r in raw]
        lraw = [rng.uniform(0.4, 1.2) for _ in range(num_layer)]
        ltot = sum(lraw)
        demands = [v / ltot for v in lraw]
        cases.append([num_stage, num_layer, caps, demands])
    ref = probe("compute_balancer", {"cases": cases})
    for case, (ref_partition, ref_demand) in zip(cases, ref):
        num_stage, num_layer, caps, demands = case
        bal = LayerComputeBalancer(num_stage, num_layer, list(caps), list(demands))
        partition, demand = bal.run()
        assert partition == ref_partition, case
        assert demand == pytest.approx(ref_demand), case


def test_homo_costs_parity(tmp_path):
    from metis_amd.cluster import ClusterSpec
    from metis_amd.config import ModelConfig, PlannerArgs
    from metis_amd.cli.homo_cluster import search_homo_cluster
    from metis_amd.profiles import ProfileStore

    hf = tmp_path / "hostfile"
    hf.write_text("n1 slots=4\nn2 slots=4\n")
    cf = tmp_path / "clusterfile.json"
    # inter == intra neutralizes the reference's Q4 getter bug
    cf.write_text(json.dumps({
        "n1": {"instance_type": "A100", "inter_bandwidth": 50,
               "intra_bandwidth": 50, "memory": 80},
        "n2": {"instance_type": "A100", "inter_bandwidth": 50,
               "intra_bandwidth": 50, "memory": 80},
    }))
    model = dict(model_name="GPT", num_layers=10, hidden_size=4096,
                 sequence_length=1024, vocab_size=51200, attention_head_size=128)
    ref = probe("homo_costs", {
        "hostfile": str(hf), "clusterfile": str(cf), "profile_dir": SAMPLES,
        "gbs": 16, "max_tp": 4, "model": model,
    })
    model_from = ref["model_file_order"][0]

    cluster = ClusterSpec(str(hf), str(cf))
    store = ProfileStore.load_dir(SAMPLES, model_from=model_from)
    cfg = ModelConfig(**{k: v for k, v in model.items()})
    results = search_homo_cluster(
        cluster, store, cfg,
        PlannerArgs(gbs=16, max_profiled_tp_degree=4, max_profiled_batch_size=16),
        device_type="A100",
    )
    mine = sorted((p.dp, p.pp, p.tp, p.mbs, round(c, 9)) for p, c, _ in results)
    theirs = sorted((r[0], r[1], r[2], r[3], round(r[4], 9)) for r in ref["rows"])
    assert mine == theirs


def test_het_search_parity_64_plans(tmp_path):
    """Reproduces the survey's verified run: 64 costed plans on the bundled
    A100 samples, emulated 8xA100 over 2 nodes, gbs=16 — every (cost, plan)
    pair bit-identical to a live run of the reference CLI."""
    from metis_amd.cluster import ClusterSpec
    from metis_amd.config import ModelConfig, PlannerArgs
    from metis_amd.cli.het_cluster import search_het_cluster
    from metis_amd.profiles import ProfileStore

    hf = tmp_path / "hostfile"
    hf.write_text("n1 slots=4\nn2 slots=4\n")
    cf = tmp_path / "clusterfile.json"
    cf.write_text(json.dumps({
        "n1": {"instance_type": "A100", "inter_bandwidth": 50,
               "intra_bandwidth": 50, "memory": 80},
        "n2": {"instance_type": "A100", "inter_bandwidth": 50,
               "intra_bandwidth": 50, "memory": 80},
    }))
    common = [
        "--model_name", "GPT", "--num_layers", "10", "--gbs", "16",
        "--hidden_size", "4096", "--sequence_length", "1024",
        "--vocab_size", "51200", "--attention_head_size", "128",
        "--hostfile_path", str(hf), "--clusterfile_path", str(cf),
        "--profile_data_path", SAMPLES,
        "--max_profiled_tp_degree", "4", "--max_profiled_batch_size", "4",
        "--min_group_scale_variance", "1", "--max_permute_len", "4",
    ]
    ref_out = subprocess.run(
        [sys.executable, "cost_het_cluster.py"] + common,
        cwd="/root/reference", capture_output=True, text=True, check=True,
    ).stdout
    ref_rows = []
    for line in ref_out.splitlines():
        parts = line.split(", ")
        if parts and parts[0].isdigit():
            # cost, device_groups..., strategies..., batches, layer_partition
            ref_rows.append((round(float(parts[1]), 8), ", ".join(parts[3:])))
    assert len(ref_rows) == 64

    model_from = probe("homo_costs", {
        "hostfile": str(hf), "clusterfile": str(cf), "profile_dir": SAMPLES,
        "gbs": 16, "max_tp": 4,
        "model": dict(model_name="GPT", num_layers=10, hidden_size=4096,
                      sequence_length=1024, vocab_size=51200,
                      attention_head_size=128),
    })["model_file_order"][0]

    cluster = ClusterSpec(str(hf), str(cf))
    store = ProfileStore.load_dir(SAMPLES, model_from=model_from)
    cfg = ModelConfig("GPT", 10, 4096, 1024, 51200, 128)
    results = search_het_cluster(
        cluster, store, cfg,
        PlannerArgs(gbs=16, max_profiled_tp_degree=4, max_profiled_batch_size=4,
                    min_group_scale_variance=1, max_permute_len=4),
    )
    assert len(results) == 64
    mine = sorted(
        (round(r[6], 8), f"{r[1]}, {r[2]}, {r[3]}, {r[4]}") for r in results
    )
    assert mine == sorted(ref_rows)


def test_het_search_parity_two_device_types(tmp_path):
    """Hetero parity on a GENUINELY mixed cluster (2 device types): covers
    DataLoadBalancer, the hetero execution-cost path, HetClusterBandwidth
    and node-sequence placement against a live reference run. Device names
    A100/T4 because the reference's DeviceType enum is closed."""
    import importlib.util

    spec = importlib.util.spec_from_file_location(
        "gen_synth", os.path.join(REPO, "scripts", "gen_synth_profiles.py"))
    gen = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(gen)
    prof = tmp_path / "prof"
    gen.main(str(prof), device_speeds=(("A100", 1.0), ("T4", 0.4)))

    hf = tmp_path / "hostfile"
    hf.write_text("n1 slots=4\nn2 slots=4\n")
    cf = tmp_path / "clusterfile.json"
    # per-node inter == intra neutralizes reference quirk Q4
    cf.write_text(json.dumps({
        "n1": {"instance_type": "A100", "inter_bandwidth": 60,
               "intra_bandwidth": 60, "memory": 80},
        "n2": {"instance_type": "T4", "inter_bandwidth": 40,
               "intra_bandwidth": 40, "memory": 16},
    }))
    common = [
        "--model_name", "GPT", "--num_layers", "10", "--gbs", "16",
        "--hidden_size", "4096", "--sequence_length", "1024",
        "--vocab_size", "51200", "--attention_head_size", "128",
        "--hostfile_path", str(hf), "--clusterfile_path", str(cf),
        "--profile_data_path", str(prof),
        "--max_profiled_tp_degree", "4", "--max_profiled_batch_size", "4",
        "--min_group_scale_variance", "1", "--max_permute_len", "4",
    ]
    ref_out = subprocess.run(
        [sys.executable, "cost_het_cluster.py"] + common,
        cwd="/root/reference", capture_output=True, text=True, check=True,
    ).stdout
    import re as _re

    ref_rows = []
    for line in ref_out.splitlines():
        parts = line.split(", ")
        if not (parts and parts[0].isdigit()):
            continue
        # node_sequence itself contains commas: compare the row from the
        # device_groups bracket onward. The reference enumerates
        # permutations(set(...)) whose ORDER is per-process random (enum
        # identity hashing), and quirk Q12 ties which sequence carries the
        # single-stage plans to that order — so the sequence label is not
        # comparable, while costs and plan content are.
        rest = line[line.index(", [") + 2:]
        ref_rows.append((round(float(parts[1]), 8), rest))
    assert ref_rows, ref_out[-2000:]
    # quirk Q12 re-enumerates the SECOND node sequence's first multi-stage
    # group list, and WHICH sequence is second is the reference's random
    # set order — compare unique plans
    ref_rows = sorted(set(ref_rows))

    model_from = probe("homo_costs", {
        "hostfile": str(hf), "clusterfile": str(cf), "profile_dir": str(prof),
        "gbs": 16, "max_tp": 4,
        "model": dict(model_name="GPT", num_layers=10, hidden_size=4096,
                      sequence_length=1024, vocab_size=51200,
                      attention_head_size=128),
    })["model_file_order"][0]

    from metis_amd.cluster import ClusterSpec, device_registry
    from metis_amd.config import ModelConfig, PlannerArgs
    from metis_amd.cli.het_cluster import search_het_cluster
    from metis_amd.profiles import ProfileStore
    import metis_amd.cli.het_cluster as het_mod

    cluster = ClusterSpec(str(hf), str(cf))
    store = ProfileStore.load_dir(str(prof), model_from=model_from)

    # The reference permutes a SET of device types, whose order is
    # per-process random — and quirk Q12 treats the 2nd+ sequence
    # differently, so the costed set depends on that order. Align by
    # parsing the order the reference actually used from its debug output.
    import re as _re

    m = _re.search(r"node_sequence=\(<DeviceType\.(\w+):.*?<DeviceType\.(\w+):",
                   ref_out)
    assert m, "could not parse the reference's first node sequence"
    seq_names = [m.group(1).upper(), m.group(2).upper()]
    ordered_types = [device_registry.get(n) for n in seq_names]

    from metis_amd.planner.inter_stage import inter_stage_plans
    from metis_amd.planner.balancer import LayerLoadBalancer, StagePerformance
    from metis_amd.planner.intra_stage import intra_stage_plans
    from metis_amd.planner.cost import HeteroCostEstimator
    from metis_amd.planner.volume import GPTVolume

    cfg = ModelConfig("GPT", 10, 4096, 1024, 51200, 128)
    pa = PlannerArgs(gbs=16, max_profiled_tp_degree=4, max_profiled_batch_size=4,
                     min_group_scale_variance=1, max_permute_len=4)
    volume = GPTVolume(cfg, store.model.parameters_per_layer_bytes,
                       pa.activation_dtype_bytes)
    estimator = HeteroCostEstimator(store, cfg, volume, cluster, pa)
    balancer = LayerLoadBalancer(cluster, store, cfg, pa.gbs)
    results = []
    for inter_plan in inter_stage_plans(ordered_types, cluster.total_devices,
                                        pa.gbs, cfg.num_layers,
                                        pa.min_group_scale_variance,
                                        pa.max_permute_len):
        stage_perf = StagePerformance(cfg, store, cluster, inter_plan)
        try:
            for intra in intra_stage_plans(inter_plan, stage_perf, balancer,
                                           pa.max_profiled_tp_degree,
                                           pa.max_profiled_batch_size):
                try:
                    cost = estimator.get_cost(inter_plan, intra.strategies,
                                              intra.layer_partition,
                                              stage_perf.rank_device_map)
                except KeyError:
                    continue
                results.append((tuple(inter_plan.node_sequence),
                                list(inter_plan.device_groups),
                                list(intra.strategies), inter_plan.batches,
                                list(intra.layer_partition),
                                intra.num_repartition, cost))
        except KeyError:
            continue
    mine = sorted({
        (round(r[6], 8), f"{r[1]}, {r[2]}, {r[3]}, {r[4]}") for r in results
    })
    theirs = ref_rows
    assert len(mine) == len(theirs), (len(mine), len(theirs))
    assert mine == theirs


def test_het_search_parity_randomized_configs(tmp_path):
    """Randomized-config live parity: single-type clusters with varied
    gbs / node count / slots / bandwidth / synthetic profile speeds, full
    search compared cost-and-plan-exact against the reference CLI."""
    import importlib.util

    spec = importlib.util.spec_from_file_location(
        "gen_synth2", os.path.join(REPO, "scripts", "gen_synth_profiles.py"))
    gen = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(gen)

    from metis_amd.cluster import ClusterSpec
    from metis_amd.config import ModelConfig, PlannerArgs
    from metis_amd.cli.het_cluster import search_het_cluster
    from metis_amd.profiles import ProfileStore

    rng = random.Random(20260913)
    for case in range(3):
        prof = tmp_path / f"prof{case}"
        speed = rng.choice([0.6, 1.0, 1.4])
        gen.main(str(prof), device_speeds=(("A100", speed),))
        nodes = rng.choice([2, 4])
        slots = rng.choice([2, 4])
        bw = rng.choice([25, 50, 100])
        gbs = rng.choice([8, 16, 32])
        hf = tmp_path / f"hostfile{case}"
        hf.write_text("".join(f"n{i} slots={slots}\n" for i in range(nodes)))
        cf = tmp_path / f"clusterfile{case}.json"
        cf.write_text(json.dumps({
            f"n{i}": {"instance_type": "A100", "inter_bandwidth": bw,
                      "intra_bandwidth": bw, "memory": 80}
            for i in range(nodes)
        }))
        common = [
            "--model_name", "GPT", "--num_layers", "10", "--gbs", str(gbs),
            "--hidden_size", "4096", "--sequence_length", "1024",
            "--vocab_size", "51200", "--attention_head_size", "128",
            "--hostfile_path", str(hf), "--clusterfile_path", str(cf),
            "--profile_data_path", str(prof),
            "--max_profiled_tp_degree", "4", "--max_profiled_batch_size", "4",
            "--min_group_scale_variance", "1", "--max_permute_len", "4",
        ]
        ref_out = subprocess.run(
            [sys.executable, "cost_het_cluster.py"] + common,
            cwd="/root/reference", capture_output=True, text=True, check=True,
        ).stdout
        ref_rows = []
        for line in ref_out.splitlines():
            parts = line.split(", ")
            if parts and parts[0].isdigit():
                rest = line[line.index(", [") + 2:]
                ref_rows.append((round(float(parts[1]), 8), rest))

        model_from = probe("homo_costs", {
            "hostfile": str(hf), "clusterfile": str(cf),
            "profile_dir": str(prof), "gbs": gbs, "max_tp": 4,
            "model": dict(model_name="GPT", num_layers=10, hidden_size=4096,
                          sequence_length=1024, vocab_size=51200,
                          attention_head_size=128),
        })["model_file_order"][0]
        cluster = ClusterSpec(str(hf), str(cf))
        store = ProfileStore.load_dir(str(prof), model_from=model_from)
        cfg = ModelConfig("GPT", 10, 4096, 1024, 51200, 128)
        results = search_het_cluster(
            cluster, store, cfg,
            PlannerArgs(gbs=gbs, max_profiled_tp_degree=4,
                        max_profiled_batch_size=4,
                        min_group_scale_variance=1, max_permute_len=4))
        mine = sorted(
            (round(r[6], 8),
             f"{r[1]}, {r[2]}, {r[3]}, {r[4]}") for r in results)
        assert mine == sorted(ref_rows), (
            case, speed, nodes, slots, bw, gbs, len(mine), len(ref_rows))


def test_data_balancer_parity_fuzzed():
    """DataLoadBalancer vs the reference on random hetero DP groups
    (largest-remainder rounding, per-type throughput bias)."""
    from metis_amd.planner.balancer import DataLoadBalancer
    from metis_amd.profiles import LayerProfile, ProfileStore

    rng = random.Random(99)
    types = ["A100", "V100", "P100", "T4"]
    cases = []
    for _ in range(30):
        dp = rng.choice([2, 3, 4])
        group = rng.choice([1, 2])
        seq = [rng.choice(types) for _ in range(dp)]
        device_types = [t for t in seq for _ in range(group)]
        bs = rng.randint(dp, 64)
        totals = {t: round(rng.uniform(5.0, 50.0), 3) for t in set(seq)}
        cases.append([device_types, dp, 1, bs, totals])
    ref = probe("data_balancer", {"cases": cases})

    for case, expected in zip(cases, ref):
        device_types, dp, tp, bs, totals = case
        store = ProfileStore.__new__(ProfileStore)
        store._data = {
            (t, tp, 1): LayerProfile(layer_times_ms=[v],
                                     layer_memory_mb=[0.0], fb_sync_ms=0.0)
            for t, v in totals.items()
        }
        mine = DataLoadBalancer(store).partition_data(device_types, (dp, tp), bs)
        assert mine == expected, case
