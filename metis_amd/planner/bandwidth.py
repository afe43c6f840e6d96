"""Cluster communication topology: slowest-link bandwidth per group.

Parity: reference model/cluster_bandwidth.py:71-195, with the reference's
inter/intra getter bug fixed (gpu_cluster.py:56-58 returns intra for both —
quirk Q4; here inter means inter).

For MI355X the clusterfile's ``intra_bandwidth`` should carry the
*measured* RCCL all-reduce bus bandwidth over the xGMI mesh (a ring
all-reduce is single-link bound: ~153 GB/s per hop out of 7 p2p links per
GPU — SURVEY.md §5.8), and ``inter_bandwidth`` the measured NIC bandwidth.
``metis_amd.profiler.comm_bench`` produces both.
"""

from __future__ import annotations

from collections import Counter
from typing import List, Sequence, Tuple

from metis_amd.cluster import ClusterSpec
from metis_amd.planner.plans import InterStagePlan


class HomoTopology:
    """Uniform plans: rank grid (pp, dp, tp), rank -> node by division."""

    def __init__(self, cluster: ClusterSpec) -> None:
        self.cluster = cluster
        self.total = cluster.total_devices
        self.per_node = cluster.devices_per_node
        self.intra = cluster.intra_bandwidth(0)
        self.inter = cluster.inter_bandwidth(0)

    def _same_node(self, ranks: Sequence[int]) -> bool:
        nodes = {r // self.per_node for r in ranks}
        return len(nodes) == 1

    def slowest_pp_bandwidth(self, strategy: Tuple[int, int, int], stage_id: int) -> float:
        """Slowest link among the p2p pairs between stage_id and stage_id+1.

        Rank layout: rank = (p * dp + d) * tp + t (grid reshape(pp, dp, tp),
        parity with cluster_bandwidth.py:83-100).
        """
        pp, tp, dp = strategy
        assert pp * tp * dp == self.total, "uniform strategy must cover the cluster"
        assert stage_id < pp, "stage_id cannot be greater than pp_deg"
        bw = self.intra
        for d in range(dp):
            for t in range(tp):
                a = (stage_id * dp + d) * tp + t
                b = ((stage_id + 1) * dp + d) * tp + t
                if not self._same_node((a, b)):
                    bw = self.inter
        return bw

    def slowest_dp_bandwidth(self, strategy: Tuple[int, int, int]) -> float:
        """Slowest class over each stage's full rank block (parity:
        cluster_bandwidth.py:102-132 — the reference's dp 'groups' are the
        per-stage flattened rank sets)."""
        pp, tp, dp = strategy
        assert pp * tp * dp == self.total, "uniform strategy must cover the cluster"
        bw = self.intra
        stage_size = dp * tp
        for p in range(pp):
            ranks = range(p * stage_size, (p + 1) * stage_size)
            if not self._same_node(ranks):
                bw = self.inter
        return bw


class HeteroTopology:
    """Non-uniform plans: stages own contiguous rank ranges; device types
    follow the plan's node sequence."""

    def __init__(self, cluster: ClusterSpec, plan: InterStagePlan) -> None:
        self.cluster = cluster
        self.plan = plan
        self.per_node = cluster.devices_per_node
        # per-node device-type names, reordered by the plan's node sequence
        type_counts = Counter(t.name for t in cluster.device_types())
        self.node_types: List[str] = []
        for spec in plan.node_sequence:
            self.node_types.extend([spec.name] * type_counts[spec.name])

    def _stage_ranks(self, stage_id: int, n_stages: int = 1) -> List[int]:
        groups = self.plan.device_groups
        start = sum(groups[:stage_id])
        end = sum(groups[: stage_id + n_stages])
        return list(range(start, end))

    def _group_bandwidth(self, ranks: Sequence[int]) -> float:
        """Intra-node class only when the group sits on ONE node; otherwise
        the slowest inter bandwidth among nodes of the involved types
        (parity: cluster_bandwidth.py:169-195 — its per-unique-node type
        list has length 1 exactly when one node is involved)."""
        node_ids = {r // self.per_node for r in ranks}
        types = {self.node_types[n] for n in node_ids}
        if len(node_ids) == 1:
            return self.cluster.intra_bandwidth_for_type(next(iter(types)))
        return self.cluster.min_inter_bandwidth_for_types(types)

    def slowest_pp_bandwidth(self, stage_id: int) -> float:
        return self._group_bandwidth(self._stage_ranks(stage_id, n_stages=2))

    def slowest_dp_bandwidth(self, strategy: Tuple[int, int], stage_id: int) -> float:
        dp_deg, tp_deg = strategy
        ranks = self._stage_ranks(stage_id)
        # round-robin deal over tp first (parity: cluster_bandwidth.py:148-156)
        dp_groups: List[List[int]] = [[] for _ in range(dp_deg)]
        it = iter(ranks)
        for _t in range(tp_deg):
            for d in range(dp_deg):
                dp_groups[d].append(next(it))
        return min(self._group_bandwidth(g) for g in dp_groups)
