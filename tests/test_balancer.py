import pytest

from metis_amd.cluster import ClusterSpec
from metis_amd.config import ModelConfig
from metis_amd.planner.balancer import (
    DataLoadBalancer,
    LayerComputeBalancer,
    LayerLoadBalancer,
    StagePerformance,
    pow2_slices,
)
from metis_amd.planner.plans import InterStagePlan
from metis_amd.profiles import ProfileStore


def test_pow2_slices():
    assert pow2_slices(5) == [4, 1]
    assert pow2_slices(7) == [4, 2, 1]
    assert pow2_slices(8) == [8]
    assert pow2_slices(1) == [1]
    assert pow2_slices(0) == []


@pytest.fixture(scope="module")
def store(sample_profile_dir):
    # module-scoped: see conftest sample_profile_dir
    return ProfileStore.load_dir(str(sample_profile_dir))


@pytest.fixture()
def het_cluster(tmp_path):
    import json

    (tmp_path / "hostfile").write_text("h1 slots=8\nh2 slots=8\n")
    (tmp_path / "clusterfile.json").write_text(json.dumps({
        "h1": {"instance_type": "MI355X", "inter_bandwidth": 40,
               "intra_bandwidth": 130, "memory": 288},
        "h2": {"instance_type": "MI355X_LC", "inter_bandwidth": 40,
               "intra_bandwidth": 110, "memory": 288},
    }))
    return ClusterSpec(str(tmp_path / "hostfile"), str(tmp_path / "clusterfile.json"))


def test_data_load_balancer_sums_and_bias(store):
    balancer = DataLoadBalancer(store)
    # 2 replicas: one fast (MI355X), one 2x slower (MI355X_LC)
    types = ["MI355X", "MI355X_LC"]
    alloc = balancer.partition_data(types, (2, 1), 12)
    assert sum(alloc) == 12
    assert alloc[0] > alloc[1]  # fast device gets more


def test_compute_balancer_partition_covers_layers():
    layers = 10
    norm = [0.05] + [0.1] * 8 + [0.15]
    for caps in ([0.5, 0.5], [0.7, 0.3], [0.25, 0.25, 0.5], [1.0]):
        bal = LayerComputeBalancer(len(caps), layers, list(caps), norm)
        partition, demand = bal.run()
        assert partition[0] == 0
        assert partition[-1] == layers
        assert len(partition) == len(caps) + 1
        assert all(b >= a for a, b in zip(partition, partition[1:]))
        assert len(demand) == len(caps)


def test_compute_balancer_proportionality():
    # a 3x faster stage should receive more layers
    norm = [0.1] * 10
    bal = LayerComputeBalancer(2, 10, [0.75, 0.25], norm)
    partition, _ = bal.run()
    assert partition[1] >= 6


def _inter_plan(cluster, groups, batches=4, gbs=16):
    return InterStagePlan(
        ns_idx=0,
        node_sequence=cluster.unique_device_types(),
        dg_idx=0,
        device_groups=groups,
        num_stage=len(groups),
        batches=batches,
        gbs=gbs,
    )


def test_stage_performance_normalized(store, het_cluster):
    cfg = ModelConfig("GPT", 10, 4096, 1024, 51200)
    plan = _inter_plan(het_cluster, [8, 8])
    perf = StagePerformance(cfg, store, het_cluster, plan)
    compute = perf.compute_performance([(4, 2), (4, 2)], 16, 4)
    assert sum(compute) == pytest.approx(1.0)
    # the MI355X stage (2x faster in synth data) outperforms the LC stage
    assert compute[0] > compute[1]
    caps = perf.memory_capacity()
    assert caps == [288 * 1024 * 8, 288 * 1024 * 8]


def test_layer_load_balancer_partition(store, het_cluster):
    cfg = ModelConfig("GPT", 10, 4096, 1024, 51200)
    plan = _inter_plan(het_cluster, [8, 8])
    perf = StagePerformance(cfg, store, het_cluster, plan)
    llb = LayerLoadBalancer(het_cluster, store, cfg, 16, norm_device_type="MI355X")
    strategies = [(4, 2), (4, 2)]
    compute = perf.compute_performance(strategies, 16, 4)
    partition, attempts, state = llb.partition_layer(
        plan, strategies, compute, perf.memory_capacity()
    )
    assert partition is not None
    assert partition[0] == 0 and partition[-1] == 10
    assert attempts >= 1
    assert len(state) == 2
    # plenty of memory (288 GB pooled) -> fits on the first attempt
    assert attempts == 1


def test_layer_load_balancer_oom_path(store, het_cluster):
    cfg = ModelConfig("GPT", 10, 4096, 1024, 51200)
    plan = _inter_plan(het_cluster, [8, 8])
    perf = StagePerformance(cfg, store, het_cluster, plan)
    llb = LayerLoadBalancer(het_cluster, store, cfg, 16, norm_device_type="MI355X")
    strategies = [(4, 2), (4, 2)]
    compute = perf.compute_performance(strategies, 16, 4)
    # absurdly small capacity: must give up, not loop forever
    partition, attempts, state = llb.partition_layer(
        plan, strategies, compute, [1.0, 1.0]
    )
    assert partition is None
    assert attempts == -1
