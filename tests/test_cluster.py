import json
import os

import pytest

from metis_amd.cluster import ClusterSpec, DeviceSpec, device_registry, parse_hostfile


def _write_cluster(tmp_path, hosts, info):
    hostfile = tmp_path / "hostfile"
    hostfile.write_text("".join(hosts))
    clusterfile = tmp_path / "clusterfile.json"
    clusterfile.write_text(json.dumps(info))
    return str(hostfile), str(clusterfile)


def test_hostfile_both_formats(tmp_path):
    p = tmp_path / "hf"
    p.write_text("10.0.0.1 slots=16\n10.0.0.2 8\n\n# comment\n10.0.0.3 slots=4\n")
    entries = parse_hostfile(str(p))
    assert [e["num_device"] for e in entries] == [16, 8, 4]
    assert entries[0]["ip"] == "10.0.0.1"


def test_hostfile_multidigit_slots(tmp_path):
    # the reference reads a single char (quirk Q5); we must not
    p = tmp_path / "hf"
    p.write_text("a.b.c.d slots=12\n")
    assert parse_hostfile(str(p))[0]["num_device"] == 12


def test_registry_open_types():
    spec = device_registry.get("SOME_FUTURE_GPU")
    assert spec.name == "SOME_FUTURE_GPU"
    assert device_registry.get("MI355X").memory_gb == 288.0
    assert str(device_registry.get("MI355X")) == "DeviceType.MI355X"


def test_cluster_spec(tmp_path):
    hf, cf = _write_cluster(
        tmp_path,
        ["1.1.1.1 slots=8\n", "1.1.1.2 slots=8\n"],
        {
            "1.1.1.1": {"instance_type": "MI355X", "inter_bandwidth": 40,
                        "intra_bandwidth": 130, "memory": 288},
            "1.1.1.2": {"instance_type": "MI355X_LC", "inter_bandwidth": 35,
                        "intra_bandwidth": 110, "memory": 288},
        },
    )
    c = ClusterSpec(hf, cf)
    assert c.total_devices == 16
    assert c.devices_per_node == 8
    assert c.num_nodes == 2
    assert c.device_memory_mb(0) == 288 * 1024
    assert c.num_devices_of_type("MI355X") == 8
    assert c.num_nodes_of_type("MI355X_LC") == 1
    assert c.intra_bandwidth_for_type("MI355X_LC") == 110
    # inter is genuinely inter (reference quirk Q4 fixed)
    assert c.min_inter_bandwidth_for_types(["MI355X", "MI355X_LC"]) == 35
    assert [t.name for t in c.unique_device_types()] == ["MI355X", "MI355X_LC"]


def test_cluster_missing_ip_raises(tmp_path):
    hf, cf = _write_cluster(
        tmp_path, ["9.9.9.9 slots=4\n"],
        {"1.1.1.1": {"instance_type": "MI355X", "inter_bandwidth": 40,
                     "intra_bandwidth": 130, "memory": 288}},
    )
    with pytest.raises(KeyError):
        ClusterSpec(hf, cf)
