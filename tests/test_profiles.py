import json
import os

import pytest

from metis_amd.profiles import ProfileStore


def test_load_synth(sample_profile_dir):
    store = ProfileStore.load_dir(sample_profile_dir)
    assert set(store.device_type_names) == {"MI355X", "MI355X_LC"}
    assert store.model.num_layers == 10
    prof = store.get("MI355X", 1, 1)
    assert len(prof.layer_times_ms) == 10
    assert len(prof.layer_memory_mb) == 10
    # fb_sync is the residual: fwd_bwd - sum(layer times); synth uses 8%
    assert prof.fb_sync_ms == pytest.approx(sum(prof.layer_times_ms) * 0.08, rel=1e-6)


def test_optimizer_time_doubled(sample_profile_dir, tmp_path):
    # synth writes optimizer_time_ms = 10.0 for MI355X tp1 (sorted-first file
    # is MI355X_LC tp1_bs1 with optimizer = 10/1/0.5 = 20 -> doubled 40)
    store = ProfileStore.load_dir(sample_profile_dir)
    first = sorted(os.listdir(sample_profile_dir))[0]
    with open(os.path.join(sample_profile_dir, first)) as fh:
        raw = json.load(fh)
    assert store.model.optimizer_time_ms == raw["execution_time"]["optimizer_time_ms"] * 2


def test_missing_point_raises(sample_profile_dir):
    store = ProfileStore.load_dir(sample_profile_dir)
    with pytest.raises(KeyError):
        store.get("MI355X", 8, 1)
    with pytest.raises(KeyError):
        store.get("MI355X", 1, 3)


def test_fb_sync_zero_treated_missing(sample_profile_dir):
    store = ProfileStore.load_dir(sample_profile_dir)
    store.get("MI355X", 1, 1).fb_sync_ms = 0.0
    with pytest.raises(KeyError):
        store.fb_sync("MI355X", 1, 1)


def test_write_read_roundtrip(tmp_path):
    path = tmp_path / "DeviceType.MI355X_tp2_bs4.json"
    ProfileStore.write_profile_json(
        str(path),
        model_name="gpt-test",
        parameters_per_layer_bytes=[10.0, 20.0, 10.0],
        total_time_ms=100.0,
        forward_backward_time_ms=90.0,
        batch_generator_time_ms=1.0,
        layernorm_grads_all_reduce_time_ms=0.5,
        embedding_grads_all_reduce_time_ms=0.7,
        optimizer_time_ms=5.0,
        layer_compute_total_ms=[10.0, 60.0, 15.0],
        total_memory_mb=300.0,
        layer_memory_total_mb=[50.0, 200.0, 50.0],
    )
    store = ProfileStore.load_dir(str(tmp_path))
    prof = store.get("MI355X", 2, 4)
    assert prof.layer_times_ms == [10.0, 60.0, 15.0]
    assert prof.fb_sync_ms == pytest.approx(90.0 - 85.0)
    assert store.model.optimizer_time_ms == 10.0
    assert store.model.parameters_per_layer_bytes == [10.0, 20.0, 10.0]


def test_emulated_tp_profile_ingest(tmp_path):
    """Profiles with the tp_comm_modeled extension keys load like any
    other (the loader keeps unknown execution_time keys harmless) and
    still expose the accumulation marginal."""
    from metis_amd.profiles import ProfileStore

    path = tmp_path / "DeviceType.MI355X_tp2_bs1.json"
    ProfileStore.write_profile_json(
        str(path), model_name="m",
        parameters_per_layer_bytes=[10.0, 20.0, 10.0],
        total_time_ms=12.0, forward_backward_time_ms=10.0,
        batch_generator_time_ms=0.1,
        layernorm_grads_all_reduce_time_ms=0.3,
        embedding_grads_all_reduce_time_ms=0.2,
        optimizer_time_ms=1.0,
        layer_compute_total_ms=[2.0, 5.0, 2.0],
        total_memory_mb=30.0, layer_memory_total_mb=[10.0, 10.0, 10.0],
        fwd_bwd_2mb_ms=18.0, fwd_bwd_4mb_ms=34.0,
        extra_execution_keys={"tp_comm_modeled": True,
                              "tp_comm_bw_GBps": 130.0,
                              "tp_comm_alpha_us": 20.0})
    store = ProfileStore.load_dir(str(tmp_path))
    prof = store.get("MI355X", 2, 1)
    assert prof.marginal_mb_ms == 8.0      # (34-18)/2
    assert prof.residual_ms == 2.0         # 18 - 2*8
    import json
    raw = json.load(open(path))
    assert raw["execution_time"]["tp_comm_modeled"] is True
