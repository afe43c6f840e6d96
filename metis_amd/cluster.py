"""Cluster description: device registry, hostfile/clusterfile parsing.

Reference parity: /root/reference/gpu_cluster.py:8-58 and utils.py:8-85,
re-designed for MI355X:

* ``DeviceType`` is an *open registry* (the reference hardcodes a closed
  enum {A100, V100, P100, T4}, utils.py:46-57) — MI355X plus clock-capped
  variants (``MI355X_LC``) register themselves, and unknown types coming
  from a clusterfile are auto-registered so profiles drive the planner,
  not a hardcoded list.
* The hostfile parser accepts both ``IP slots=N`` and ``IP N`` with
  multi-digit counts (the reference reads a single character,
  utils.py:15 — quirk Q5 in SURVEY.md Appendix C).
* ``inter_bandwidth`` is honoured (the reference's getter returns
  intra_bandwidth, gpu_cluster.py:56-58 — quirk Q4). Tests that need
  bit-parity with the reference use clusterfiles with inter == intra.
"""

from __future__ import annotations

import json
import re
from dataclasses import dataclass
from typing import Dict, List


@dataclass(frozen=True)
class DeviceSpec:
    """A GPU device type known to the planner."""

    name: str                      # e.g. "MI355X"
    memory_gb: float = 0.0         # per-GPU HBM capacity (clusterfile overrides)
    peak_bf16_tflops: float = 0.0  # dense MFMA peak, informational

    def __str__(self) -> str:  # printed inside plans, keep it short
        return f"DeviceType.{self.name}"

    def __repr__(self) -> str:
        return str(self)


class DeviceRegistry:
    """Open set of device types (replaces the reference's closed enum)."""

    def __init__(self) -> None:
        self._types: Dict[str, DeviceSpec] = {}

    def register(self, spec: DeviceSpec) -> DeviceSpec:
        self._types[spec.name.upper()] = spec
        return spec

    def get(self, name: str) -> DeviceSpec:
        """Look up (auto-registering unknown names, so clusterfiles rule)."""
        key = name.upper()
        if key not in self._types:
            self._types[key] = DeviceSpec(name=key)
        return self._types[key]

    def known(self) -> List[str]:
        return sorted(self._types)


device_registry = DeviceRegistry()

# MI355X: 288 GB HBM3E, ~2.5 PF dense bf16 MFMA (MI355X_MICROARCH.md).
device_registry.register(DeviceSpec("MI355X", memory_gb=288.0, peak_bf16_tflops=2500.0))
# Clock-capped MI355X used as the second type of the emulated hetero cluster.
device_registry.register(DeviceSpec("MI355X_LC", memory_gb=288.0, peak_bf16_tflops=1250.0))
# Reference device types so the bundled A100 sample profiles load.
for _name, _mem in (("A100", 80.0), ("V100", 32.0), ("P100", 16.0), ("T4", 16.0)):
    device_registry.register(DeviceSpec(_name, memory_gb=_mem))


_HOSTLINE = re.compile(r"^\s*(\S+)\s+(?:slots=)?(\d+)\s*$")


def parse_hostfile(path: str) -> List[dict]:
    """Parse ``IP slots=N`` / ``IP N`` lines -> [{'ip', 'num_device'}, ...].

    Multi-digit slot counts accepted (reference reads one char: quirk Q5).
    """
    entries: List[dict] = []
    with open(path, "rt") as fh:
        for line in fh:
            line = line.strip()
            if not line or line.startswith("#"):
                continue
            m = _HOSTLINE.match(line)
            if not m:
                raise ValueError(f"unparseable hostfile line: {line!r}")
            entries.append({"ip": m.group(1), "num_device": int(m.group(2))})
    return entries


def parse_clusterfile(path: str) -> Dict[str, dict]:
    """Clusterfile JSON: {ip: {instance_type, inter_bandwidth, intra_bandwidth, memory}}."""
    with open(path, "r") as fh:
        return json.load(fh)


@dataclass
class GPUNode:
    ip: str
    device_type: DeviceSpec
    num_devices: int
    memory_mb: float          # per-GPU memory in MB (clusterfile 'memory' GB * 1024)
    intra_bandwidth: float    # GB/s within the node (measured xGMI all-reduce bus BW)
    inter_bandwidth: float    # GB/s across nodes (NIC)


class ClusterSpec:
    """Parsed cluster: ordered list of nodes with device types and bandwidths.

    Parity notes (vs reference GPUCluster, gpu_cluster.py:8-58):
    * ``memory`` in the clusterfile is GB; exposed here in MB ("* 1024"
      matches gpu_cluster.py:43-50 — the reference docstring says bytes
      but the value is MB and the memory-demand comparison consumes MB).
    * ``inter_bandwidth`` is genuinely the inter-node bandwidth here.
    """

    def __init__(self, hostfile_path: str, clusterfile_path: str) -> None:
        host_entries = parse_hostfile(hostfile_path)
        info = parse_clusterfile(clusterfile_path)

        self.nodes: List[GPUNode] = []
        for entry in host_entries:
            ip = entry["ip"]
            if ip not in info:
                raise KeyError(f"hostfile ip {ip} missing from clusterfile")
            ninfo = info[ip]
            spec = device_registry.get(str(ninfo["instance_type"]))
            self.nodes.append(
                GPUNode(
                    ip=ip,
                    device_type=spec,
                    num_devices=entry["num_device"],
                    memory_mb=float(ninfo["memory"]) * 1024.0,
                    intra_bandwidth=float(ninfo["intra_bandwidth"]),
                    inter_bandwidth=float(ninfo["inter_bandwidth"]),
                )
            )
        if not self.nodes:
            raise ValueError("empty hostfile")

    # --- counts -----------------------------------------------------------
    @property
    def num_nodes(self) -> int:
        return len(self.nodes)

    @property
    def devices_per_node(self) -> int:
        return self.nodes[0].num_devices

    @property
    def total_devices(self) -> int:
        return sum(n.num_devices for n in self.nodes)

    # --- device types -----------------------------------------------------
    def device_types(self) -> List[DeviceSpec]:
        """Per-node device types, in hostfile order."""
        return [n.device_type for n in self.nodes]

    def unique_device_types(self) -> List[DeviceSpec]:
        seen: Dict[str, DeviceSpec] = {}
        for n in self.nodes:
            seen.setdefault(n.device_type.name, n.device_type)
        return list(seen.values())

    def num_devices_of_type(self, type_name: str) -> int:
        return sum(n.num_devices for n in self.nodes if n.device_type.name == type_name)

    def num_nodes_of_type(self, type_name: str) -> int:
        return sum(1 for n in self.nodes if n.device_type.name == type_name)

    # --- memory / bandwidth -----------------------------------------------
    def device_memory_mb(self, node_id: int = 0) -> float:
        return self.nodes[node_id].memory_mb

    def device_memory_mb_for_type(self, type_name: str) -> float:
        for n in self.nodes:
            if n.device_type.name == type_name:
                return n.memory_mb
        raise KeyError(type_name)

    def intra_bandwidth(self, node_id: int = 0) -> float:
        return self.nodes[node_id].intra_bandwidth

    def inter_bandwidth(self, node_id: int = 0) -> float:
        return self.nodes[node_id].inter_bandwidth

    def intra_bandwidth_for_type(self, type_name: str) -> float:
        for n in self.nodes:
            if n.device_type.name == type_name:
                return n.intra_bandwidth
        raise KeyError(type_name)

    def min_inter_bandwidth_for_types(self, type_names) -> float:
        """Slowest inter-node bandwidth among nodes of the given types."""
        bws = [
            n.inter_bandwidth
            for n in self.nodes
            if n.device_type.name in set(type_names)
        ]
        if not bws:
            raise KeyError(f"no nodes of types {type_names}")
        return min(bws)
