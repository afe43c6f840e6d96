import json

import pytest

from metis_amd.cluster import ClusterSpec
from metis_amd.config import ModelConfig, PlannerArgs
from metis_amd.planner.cost import HeteroCostEstimator, HomoCostEstimator
from metis_amd.planner.plans import InterStagePlan, UniformPlan
from metis_amd.planner.volume import GPTVolume, uniform_layer_split
from metis_amd.profiles import ProfileStore


@pytest.fixture(scope="module")
def store(sample_profile_dir):
    return ProfileStore.load_dir(str(sample_profile_dir))


@pytest.fixture()
def homo_cluster(tmp_path):
    (tmp_path / "hostfile").write_text("h1 slots=4\nh2 slots=4\n")
    (tmp_path / "clusterfile.json").write_text(json.dumps({
        "h1": {"instance_type": "MI355X", "inter_bandwidth": 40,
               "intra_bandwidth": 130, "memory": 288},
        "h2": {"instance_type": "MI355X", "inter_bandwidth": 40,
               "intra_bandwidth": 130, "memory": 288},
    }))
    return ClusterSpec(str(tmp_path / "hostfile"), str(tmp_path / "clusterfile.json"))


def _cfg():
    return ModelConfig("GPT", 10, 4096, 1024, 51200)


def test_uniform_layer_split():
    assert uniform_layer_split(10, 4) == [3, 2, 2, 3]
    assert uniform_layer_split(10, 1) == [10]
    assert uniform_layer_split(12, 5) == [3, 2, 2, 2, 3]
    assert uniform_layer_split(13, 5) == [3, 3, 2, 2, 3]
    assert sum(uniform_layer_split(34, 8)) == 34


def test_volume_math():
    cfg = _cfg()
    vol = GPTVolume(cfg, [100.0, 50.0] + [50.0] * 7 + [100.0])
    # boundary activation: bs*seq*hidden; last layer: bs*seq*vocab/tp
    assert vol.activation_size(3, 2, 1) == 2 * 1024 * 4096
    assert vol.activation_size(9, 2, 4) == 2 * 1024 * 51200 / 4
    sizes = vol.parameter_sizes(2)
    assert sizes[0] == 50.0 and sizes[-1] == 50.0 and sizes[1] == 25.0
    assert vol.stage_parameter_size(1, 0, 10) == pytest.approx(100 + 8 * 50 + 100)
    assert vol.stage_parameter_size(2, 0, 2) == pytest.approx(50 + 25)


def test_homo_cost_basics(store, homo_cluster):
    cfg = _cfg()
    vol = GPTVolume(cfg, store.model.parameters_per_layer_bytes)
    est = HomoCostEstimator(store, cfg, vol, homo_cluster,
                            PlannerArgs(gbs=16, max_profiled_tp_degree=4,
                                        max_profiled_batch_size=4))
    plan = UniformPlan(dp=8, pp=1, tp=1, mbs=2, gbs=16)
    cost, stage_mem, oom = est.get_cost(plan, "MI355X")
    assert cost > 0 and not oom
    assert len(stage_mem) == 1

    # more microbatches at same (dp, tp, mbs) can only add cost terms
    plan_pp = UniformPlan(dp=4, pp=2, tp=1, mbs=2, gbs=16)
    cost_pp, stage_mem_pp, _ = est.get_cost(plan_pp, "MI355X")
    assert len(stage_mem_pp) == 2

    # unprofiled point -> KeyError (callers skip)
    with pytest.raises(KeyError):
        est.get_cost(UniformPlan(dp=1, pp=1, tp=8, mbs=16, gbs=16), "MI355X")


def test_homo_cost_alpha_beta_adds_latency(store, homo_cluster):
    cfg = _cfg()
    vol = GPTVolume(cfg, store.model.parameters_per_layer_bytes)
    base_args = PlannerArgs(gbs=16, max_profiled_tp_degree=4, max_profiled_batch_size=4)
    ab_args = PlannerArgs(gbs=16, max_profiled_tp_degree=4, max_profiled_batch_size=4,
                          comm_model="alpha_beta", alpha_us=100.0)
    plan = UniformPlan(dp=4, pp=2, tp=1, mbs=2, gbs=16)
    c0, _, _ = HomoCostEstimator(store, cfg, vol, homo_cluster, base_args).get_cost(plan, "MI355X")
    c1, _, _ = HomoCostEstimator(store, cfg, vol, homo_cluster, ab_args).get_cost(plan, "MI355X")
    # alpha term: 1 pp boundary + 1 dp all-reduce at 0.1 ms each
    assert c1 == pytest.approx(c0 + 0.2)


def test_hetero_cost_runs(store, homo_cluster):
    cfg = _cfg()
    vol = GPTVolume(cfg, store.model.parameters_per_layer_bytes)
    est = HeteroCostEstimator(store, cfg, vol, homo_cluster,
                              PlannerArgs(gbs=16, max_profiled_tp_degree=4,
                                          max_profiled_batch_size=4))
    plan = InterStagePlan(
        ns_idx=0, node_sequence=homo_cluster.unique_device_types(),
        dg_idx=0, device_groups=[4, 4], num_stage=2, batches=4, gbs=16,
    )
    rank_map = {r: "MI355X" for r in range(8)}
    cost = est.get_cost(plan, [(2, 2), (2, 2)], [0, 5, 10], rank_map)
    assert cost > 0

    # single-stage plan == homo dp-only cost shape
    plan1 = InterStagePlan(
        ns_idx=0, node_sequence=homo_cluster.unique_device_types(),
        dg_idx=0, device_groups=[8], num_stage=1, batches=2, gbs=16,
    )
    cost1 = est.get_cost(plan1, [(8, 1)], [0, 10], rank_map)
    assert cost1 > 0


def test_json_out_cli(tmp_path, monkeypatch):
    """--json_out structured plan dump on the hetero CLI."""
    import json as _json
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = tmp_path / "plans.json"
    cmd = [sys.executable, "cost_het_cluster.py",
           "--model_name", "GPT", "--num_layers", "10", "--gbs", "16",
           "--hidden_size", "4096", "--sequence_length", "1024",
           "--vocab_size", "51200",
           "--hostfile_path", "tests/data/profiles_synth_hostfile"
           if os.path.exists(os.path.join(repo, "tests/data/profiles_synth_hostfile"))
           else "tests/data/mi355x_single_node/hostfile",
           "--clusterfile_path", "tests/data/mi355x_single_node/clusterfile.json",
           "--profile_data_path", "tests/data/profiles_synth",
           "--max_profiled_tp_degree", "4", "--max_profiled_batch_size", "4",
           "--top_k", "5", "--json_out", str(out)]
    r = subprocess.run(cmd, cwd=repo, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, r.stderr[-1500:]
    doc = _json.loads(out.read_text())
    assert doc["num_plans"] > 0
    first = doc["plans"][0]
    assert first["rank"] == 1 and first["cost_ms"] > 0
    assert sum(first["device_groups"]) == 8
